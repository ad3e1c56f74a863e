from perceiver_amd.data.text.bookcorpus import BookCorpusDataModule, BookCorpusOpenDataModule
from perceiver_amd.data.text.c4 import C4Collator, C4DataModule
from perceiver_amd.data.text.collator import (
    Collator,
    DefaultCollator,
    RandomTruncateCollator,
    TokenMaskingCollator,
    WordMaskingCollator,
)
from perceiver_amd.data.text.common import (
    CLMDataset,
    RandomShiftDataset,
    Task,
    TextDataModule,
    TextPreprocessor,
)
from perceiver_amd.data.text.enwik8 import Enwik8DataModule
from perceiver_amd.data.text.imdb import ImdbDataModule
from perceiver_amd.data.text.utils import PerceiverTokenizerUtil
from perceiver_amd.data.text.wikipedia import WikipediaDataModule
from perceiver_amd.data.text.wikitext import WikiTextDataModule
