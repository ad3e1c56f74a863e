"""MIDI ↔ event-token codec for symbolic-audio Perceiver-AR.

The Music-Transformer "performance" encoding: a piece becomes a stream over a
388-symbol alphabet —

    [0, 128)    note_on <pitch>
    [128, 256)  note_off <pitch>
    [256, 356)  time_shift of (value+1) × 10 ms
    [356, 388)  set velocity to value × 4

Sustain-pedal (CC 64) handling matches the reference codec (which follows
jason9693/midi-neural-processor): while the pedal is down, a note keeps
sounding until the pedal lifts or the same pitch is re-struck. Built on the
self-contained perceiver_amd.data.audio.midi reader/writer (no pretty_midi in
this environment); behavioral parity with the reference's
data/audio/midi_processor.py token streams.
"""
from __future__ import annotations

from multiprocessing import Pool
from pathlib import Path
from typing import List, Optional, Tuple

import numpy as np

from perceiver_amd.data.audio.midi import Instrument, MidiFile, Note

RANGE_NOTE_ON = 128
RANGE_NOTE_OFF = 128
RANGE_VEL = 32
RANGE_TIME_SHIFT = 100

# alphabet segment offsets, in stream order
START_IDX = {
    "note_on": 0,
    "note_off": RANGE_NOTE_ON,
    "time_shift": RANGE_NOTE_ON + RANGE_NOTE_OFF,
    "velocity": RANGE_NOTE_ON + RANGE_NOTE_OFF + RANGE_TIME_SHIFT,
}
_SEGMENTS = (("note_on", RANGE_NOTE_ON), ("note_off", RANGE_NOTE_OFF),
             ("time_shift", RANGE_TIME_SHIFT), ("velocity", RANGE_VEL))


class Event:
    """One alphabet symbol (type + value) with int round-tripping."""

    def __init__(self, event_type, value):
        self.type = event_type
        self.value = value

    def __repr__(self):
        return f"<Event type: {self.type}, value: {self.value}>"

    def to_int(self) -> int:
        return START_IDX[self.type] + self.value

    @staticmethod
    def from_int(int_value: int) -> "Event":
        at = int_value
        for name, width in _SEGMENTS:
            if at < width:
                return Event(name, at)
            at -= width
        return Event("velocity", at)  # defensive: clamp overflow into last segment


class SplitNote:
    """Half of a note: its on- or off-edge on the absolute timeline."""

    def __init__(self, type, time, value, velocity):
        self.type = type          # "note_on" | "note_off"
        self.time = time
        self.velocity = velocity
        self.value = value

    def __repr__(self):
        return f"<[SNote] time: {self.time} type: {self.type}, value: {self.value}, velocity: {self.velocity}>"


class SustainDownManager:
    """One pedal-down interval and the notes struck inside it.

    ``transposition_notes`` rewrites note ends right-to-left: a managed note
    sustains until the next strike of the same pitch, or at least until the
    pedal lifts.
    """

    def __init__(self, start, end):
        self.start = start
        self.end = end
        self.managed_notes: List[Note] = []
        self._note_dict = {}  # pitch -> start time of the following strike

    def add_managed_note(self, note: Note):
        self.managed_notes.append(note)

    def transposition_notes(self):
        for note in reversed(self.managed_notes):
            next_strike = self._note_dict.get(note.pitch)
            if next_strike is not None:
                note.end = next_strike
            else:
                note.end = max(self.end, note.end)
            self._note_dict[note.pitch] = note.start


# ------------------------------------------------------------------ encoding
def _pedal_intervals(ctrl_changes) -> List[SustainDownManager]:
    """CC-64 stream -> closed pedal-down intervals (value >= 64 = down)."""
    intervals: List[SustainDownManager] = []
    open_interval = None
    for ctrl in ctrl_changes:
        down = ctrl.value >= 64
        if down and open_interval is None:
            open_interval = SustainDownManager(start=ctrl.time, end=None)
        elif not down and open_interval is not None:
            open_interval.end = ctrl.time
            intervals.append(open_interval)
            open_interval = None
        elif not down and intervals:
            intervals[-1].end = ctrl.time
    return intervals


def _apply_sustain(intervals: List[SustainDownManager], notes: List[Note]) -> List[Note]:
    """Distribute notes over the pedal intervals and extend the managed ones."""
    out: List[Note] = []
    for interval in intervals:
        for idx, note in enumerate(notes):
            if note.start < interval.start:
                out.append(note)
            elif note.start > interval.end:
                notes = notes[idx:]
                interval.transposition_notes()
                break
            else:
                interval.add_managed_note(note)
    for interval in intervals:
        out += interval.managed_notes
    out.sort(key=lambda n: n.start)
    return out


def _note_edges(notes: List[Note]) -> List[SplitNote]:
    """Notes -> on/off edges, ordered by (note start, then edge time)."""
    notes.sort(key=lambda n: n.start)
    edges: List[SplitNote] = []
    for note in notes:
        edges.append(SplitNote("note_on", note.start, note.pitch, note.velocity))
        edges.append(SplitNote("note_off", note.end, note.pitch, None))
    edges.sort(key=lambda e: e.time)
    return edges


def _emit_time_shift(events: List[Event], prev_time: float, now: float) -> None:
    """Gap -> one or more time_shift symbols of <= 1 s each (10 ms units)."""
    remaining = int(round((now - prev_time) * 100))
    while remaining >= RANGE_TIME_SHIFT:
        events.append(Event("time_shift", RANGE_TIME_SHIFT - 1))
        remaining -= RANGE_TIME_SHIFT
    if remaining > 0:
        events.append(Event("time_shift", remaining - 1))


def encode_midi(midi: MidiFile) -> List[int]:
    """MidiFile -> performance-encoding token list."""
    notes: List[Note] = []
    for inst in midi.instruments:
        pedals = _pedal_intervals([c for c in inst.control_changes if c.number == 64])
        notes += _apply_sustain(pedals, inst.notes) if pedals else inst.notes

    events: List[Event] = []
    clock, prev_vel = 0.0, 0
    for edge in _note_edges(notes):
        _emit_time_shift(events, clock, edge.time)
        if edge.velocity is not None:
            binned = edge.velocity // 4
            # quirk preserved from the reference codec: the comparison is
            # against the previous edge's RAW velocity (None after a
            # note_off), so note_ons re-emit their velocity symbol almost
            # always — token parity matters more than economy here
            if prev_vel != binned:
                events.append(Event("velocity", binned))
        events.append(Event(edge.type, edge.value))
        clock = edge.time
        prev_vel = edge.velocity

    return [e.to_int() for e in events]


# ------------------------------------------------------------------ decoding
def _edges_from_events(events: List[Event]) -> List[SplitNote]:
    """Token stream -> on/off edges on an absolute timeline."""
    clock = 0.0
    velocity = 0
    edges: List[SplitNote] = []
    for ev in events:
        if ev.type == "time_shift":
            clock += (ev.value + 1) / 100
        if ev.type == "velocity":
            velocity = ev.value * 4
        elif ev.type in ("note_on", "note_off"):
            edges.append(SplitNote(ev.type, clock, ev.value, velocity))
    return edges


def _pair_edges(edges: List[SplitNote]) -> List[Note]:
    """Match each note_off to the open note_on of the same pitch; zero-length
    notes and dangling offs are dropped."""
    open_by_pitch = {}
    notes: List[Note] = []
    for edge in edges:
        if edge.type == "note_on":
            open_by_pitch[edge.value] = edge
            continue
        on = open_by_pitch.get(edge.value)
        if on is None or edge.time == on.time:
            continue
        notes.append(Note(on.velocity, edge.value, on.time, edge.time))
    return notes


def decode_midi(idx_array, file_path: Optional[str] = None) -> MidiFile:
    """Token list -> MidiFile (optionally written to disk)."""
    edges = _edges_from_events([Event.from_int(i) for i in idx_array])
    notes = _pair_edges(edges)
    notes.sort(key=lambda n: n.start)

    out = MidiFile()
    voice = Instrument(program=1, is_drum=False, name="perceiver_amd")
    voice.notes = notes
    out.instruments.append(voice)
    if file_path is not None:
        out.write(file_path)
    return out


# ------------------------------------------------------------ bulk encoding
def _encode_midi_file(file: Path) -> Optional[np.ndarray]:
    try:
        return np.array(encode_midi(MidiFile(str(file))), dtype=np.int16)
    except Exception as e:  # noqa: BLE001
        print(f"Error encoding midi file [{file}]: {e}")
        return None


def encode_midi_files(files: List[Path], num_workers: int) -> List[np.ndarray]:
    with Pool(processes=num_workers) as pool:
        encoded = list(pool.imap(_encode_midi_file, files))
    return [arr for arr in encoded if arr is not None]
