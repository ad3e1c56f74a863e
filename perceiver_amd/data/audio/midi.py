"""Minimal self-contained MIDI file reader/writer.

The environment has no pretty_midi/mido, so the symbolic-audio pipeline carries its
own Standard MIDI File implementation: enough of the spec (format 0/1, variable-length
deltas, running status, note on/off, control changes, tempo map) to read piano
corpora (GiantMIDI, Maestro) and to write generated performances back out.

API mirrors the pretty_midi subset the codec uses: ``MidiFile.instruments`` ->
``Instrument.notes`` (Note(velocity, pitch, start, end) in seconds) and
``Instrument.control_changes`` (ControlChange(number, value, time)).
"""
from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class Note:
    velocity: int
    pitch: int
    start: float
    end: float


@dataclass
class ControlChange:
    number: int
    value: int
    time: float


@dataclass
class Instrument:
    program: int = 0
    is_drum: bool = False
    name: str = ""
    notes: List[Note] = field(default_factory=list)
    control_changes: List[ControlChange] = field(default_factory=list)


class MidiFile:
    def __init__(self, path: Optional[str] = None):
        self.instruments: List[Instrument] = []
        self.ticks_per_beat = 480
        if path is not None:
            with open(path, "rb") as f:
                self._parse(f.read())

    # ------------------------------------------------------------------ parsing
    def _parse(self, data: bytes) -> None:
        if data[:4] != b"MThd":
            raise ValueError("not a MIDI file (missing MThd)")
        hlen = struct.unpack(">I", data[4:8])[0]
        fmt, ntrks, division = struct.unpack(">HHH", data[8:14])
        if division & 0x8000:
            raise ValueError("SMPTE time division not supported")
        self.ticks_per_beat = division
        pos = 8 + hlen

        raw_tracks = []
        for _ in range(ntrks):
            if data[pos: pos + 4] != b"MTrk":
                raise ValueError("bad track chunk")
            tlen = struct.unpack(">I", data[pos + 4: pos + 8])[0]
            raw_tracks.append(data[pos + 8: pos + 8 + tlen])
            pos += 8 + tlen

        tracks = [self._parse_track(t) for t in raw_tracks]

        # tempo map: (tick, us_per_beat) from every track, default 500000
        tempo_events = sorted(
            [(tick, val) for trk in tracks for (tick, kind, val) in trk if kind == "tempo"]
        )

        def tick_to_sec(tick: int) -> float:
            sec, last_tick, us = 0.0, 0, 500000
            for t, v in tempo_events:
                if t >= tick:
                    break
                sec += (t - last_tick) * us / 1e6 / self.ticks_per_beat
                last_tick, us = t, v
            return sec + (tick - last_tick) * us / 1e6 / self.ticks_per_beat

        # one Instrument per MIDI channel that has notes/ccs
        channels = {}

        def chan(c) -> Instrument:
            if c not in channels:
                channels[c] = Instrument()
            return channels[c]

        for trk in tracks:
            active = {}  # (channel, pitch) -> (start_tick, velocity)
            for tick, kind, val in trk:
                if kind == "on":
                    c, pitch, vel = val
                    active[(c, pitch)] = (tick, vel)
                elif kind == "off":
                    c, pitch = val
                    if (c, pitch) in active:
                        start_tick, vel = active.pop((c, pitch))
                        chan(c).notes.append(
                            Note(vel, pitch, tick_to_sec(start_tick), tick_to_sec(tick))
                        )
                elif kind == "cc":
                    c, number, value = val
                    chan(c).control_changes.append(ControlChange(number, value, tick_to_sec(tick)))
                elif kind == "program":
                    c, program = val
                    chan(c).program = program

        for c in sorted(channels):
            inst = channels[c]
            inst.is_drum = c == 9
            inst.notes.sort(key=lambda n: n.start)
            inst.control_changes.sort(key=lambda cc: cc.time)
            self.instruments.append(inst)

    @staticmethod
    def _read_varlen(data: bytes, pos: int):
        value = 0
        while True:
            b = data[pos]
            pos += 1
            value = (value << 7) | (b & 0x7F)
            if not b & 0x80:
                return value, pos

    def _parse_track(self, data: bytes):
        events = []
        pos, tick, status = 0, 0, 0
        while pos < len(data):
            delta, pos = self._read_varlen(data, pos)
            tick += delta
            b = data[pos]
            if b & 0x80:
                status = b
                pos += 1
            if status == 0xFF:  # meta
                mtype = data[pos]
                mlen, pos2 = self._read_varlen(data, pos + 1)
                body = data[pos2: pos2 + mlen]
                pos = pos2 + mlen
                if mtype == 0x51 and mlen == 3:
                    events.append((tick, "tempo", (body[0] << 16) | (body[1] << 8) | body[2]))
                if mtype == 0x2F:
                    break
            elif status in (0xF0, 0xF7):  # sysex
                slen, pos2 = self._read_varlen(data, pos)
                pos = pos2 + slen
            else:
                kind = status & 0xF0
                c = status & 0x0F
                if kind in (0x80, 0x90, 0xA0, 0xB0, 0xE0):
                    d1, d2 = data[pos], data[pos + 1]
                    pos += 2
                    if kind == 0x90 and d2 > 0:
                        events.append((tick, "on", (c, d1, d2)))
                    elif kind == 0x80 or (kind == 0x90 and d2 == 0):
                        events.append((tick, "off", (c, d1)))
                    elif kind == 0xB0:
                        events.append((tick, "cc", (c, d1, d2)))
                elif kind in (0xC0, 0xD0):
                    d1 = data[pos]
                    pos += 1
                    if kind == 0xC0:
                        events.append((tick, "program", (c, d1)))
                else:
                    raise ValueError(f"unexpected status byte {status:#x}")
        return events

    # ------------------------------------------------------------------ writing
    @staticmethod
    def _varlen(value: int) -> bytes:
        out = [value & 0x7F]
        value >>= 7
        while value:
            out.append(0x80 | (value & 0x7F))
            value >>= 7
        return bytes(reversed(out))

    def write(self, path: str) -> None:
        ppq, us = 480, 500000
        msgs = []  # (tick, order, bytes)
        for inst in self.instruments:
            c = 9 if inst.is_drum else 0
            msgs.append((0, 0, bytes([0xC0 | c, inst.program & 0x7F])))
            for cc in inst.control_changes:
                tick = round(cc.time * 1e6 / us * ppq)
                msgs.append((tick, 1, bytes([0xB0 | c, cc.number & 0x7F, cc.value & 0x7F])))
            for n in inst.notes:
                on_tick = round(n.start * 1e6 / us * ppq)
                off_tick = round(n.end * 1e6 / us * ppq)
                msgs.append((on_tick, 2, bytes([0x90 | c, n.pitch & 0x7F, max(1, n.velocity) & 0x7F])))
                msgs.append((off_tick, 1, bytes([0x80 | c, n.pitch & 0x7F, 0])))
        msgs.sort(key=lambda m: (m[0], m[1]))

        track = bytearray()
        track += self._varlen(0) + bytes([0xFF, 0x51, 0x03, (us >> 16) & 0xFF, (us >> 8) & 0xFF, us & 0xFF])
        last = 0
        for tick, _, msg in msgs:
            track += self._varlen(tick - last) + msg
            last = tick
        track += self._varlen(0) + bytes([0xFF, 0x2F, 0x00])

        with open(path, "wb") as f:
            f.write(b"MThd" + struct.pack(">IHHH", 6, 0, 1, ppq))
            f.write(b"MTrk" + struct.pack(">I", len(track)) + bytes(track))
