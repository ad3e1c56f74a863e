from perceiver_amd.data.audio.giantmidi_piano import GiantMidiPianoDataModule
from perceiver_amd.data.audio.maestro_v3 import MaestroV3DataModule
from perceiver_amd.data.audio.midi import ControlChange, Instrument, MidiFile, Note
from perceiver_amd.data.audio.midi_processor import decode_midi, encode_midi, encode_midi_files
from perceiver_amd.data.audio.symbolic import (
    SymbolicAudioCollator,
    SymbolicAudioDataModule,
    SymbolicAudioNumpyDataset,
)
