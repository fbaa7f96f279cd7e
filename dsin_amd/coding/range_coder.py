"""Integer range coder (arithmetic coding) for the entropy-model symbols.

The reference carries UNUSED hooks for an arithmetic coder
(/root/reference/src/probclass_imgcomp.py:361-482 — "see val.py /
bpp_helpers.py", files that do not exist there). This module completes the
loop: a byte-oriented 32-bit range coder driven by per-symbol integer
frequency tables, the same interface the reference's PredictionNetwork
exposes (`get_freqs` -> int64 frequencies >= 1).

Carry-less renormalization (Subbotin-style): the coder emits a byte whenever
the range top byte is fixed, and forces range alignment at low-range
boundaries, so encode is streaming and decode is symmetric.
"""

from __future__ import annotations

from typing import Iterable, List, Sequence

import numpy as np

_TOP = 1 << 24
_BOT = 1 << 16


class RangeEncoder:
    def __init__(self):
        self.low = 0
        self.range = 0xFFFFFFFF
        self.out = bytearray()

    def encode(self, cum: int, freq: int, tot: int) -> None:
        """Encode a symbol occupying [cum, cum+freq) of [0, tot)."""
        r = self.range // tot
        self.low = (self.low + r * cum) & 0xFFFFFFFFFFFFFFFF
        self.range = r * freq
        self._normalize()

    def _normalize(self):
        while True:
            if (self.low ^ (self.low + self.range)) < _TOP:
                pass  # top byte settled
            elif self.range < _BOT:
                self.range = (-self.low) & (_BOT - 1)
                if self.range == 0:
                    self.range = _BOT
            else:
                break
            self.out.append((self.low >> 24) & 0xFF)
            self.low = (self.low << 8) & 0xFFFFFFFF
            self.range = (self.range << 8) & 0xFFFFFFFFFFFFFFFF
            if self.range > 0xFFFFFFFF:
                self.range = 0xFFFFFFFF

    def finish(self) -> bytes:
        for _ in range(4):
            self.out.append((self.low >> 24) & 0xFF)
            self.low = (self.low << 8) & 0xFFFFFFFF
        return bytes(self.out)


class RangeDecoder:
    def __init__(self, data: bytes):
        self.data = data
        self.pos = 0
        self.low = 0
        self.range = 0xFFFFFFFF
        self.code = 0
        for _ in range(4):
            self.code = ((self.code << 8) | self._byte()) & 0xFFFFFFFF

    def _byte(self) -> int:
        b = self.data[self.pos] if self.pos < len(self.data) else 0
        self.pos += 1
        return b

    def decode_cum(self, tot: int) -> int:
        """Return the cumulative-frequency position of the next symbol."""
        self._r = self.range // tot
        cum = (self.code - self.low) // self._r
        return min(cum & 0xFFFFFFFF, tot - 1)

    def decode_update(self, cum: int, freq: int, tot: int) -> None:
        r = self._r
        self.low = (self.low + r * cum) & 0xFFFFFFFFFFFFFFFF
        self.range = r * freq
        while True:
            if (self.low ^ (self.low + self.range)) < _TOP:
                pass
            elif self.range < _BOT:
                self.range = (-self.low) & (_BOT - 1)
                if self.range == 0:
                    self.range = _BOT
            else:
                break
            self.code = ((self.code << 8) | self._byte()) & 0xFFFFFFFF
            self.low = (self.low << 8) & 0xFFFFFFFF
            self.range = (self.range << 8) & 0xFFFFFFFFFFFFFFFF
            if self.range > 0xFFFFFFFF:
                self.range = 0xFFFFFFFF


def _cumulate(freqs: np.ndarray):
    cum = np.zeros(len(freqs) + 1, dtype=np.int64)
    np.cumsum(freqs, out=cum[1:])
    return cum


def encode_with_freqs(symbols: Sequence[int],
                      freq_rows: Iterable[np.ndarray]) -> bytes:
    """Encode symbols[i] with the i-th frequency row (int >= 1 each)."""
    enc = RangeEncoder()
    for s, freqs in zip(symbols, freq_rows):
        cum = _cumulate(freqs)
        enc.encode(int(cum[s]), int(freqs[s]), int(cum[-1]))
    return enc.finish()


def decode_with_freqs(data: bytes, n: int, next_freqs) -> List[int]:
    """Decode n symbols; next_freqs(i, decoded_so_far) -> frequency row."""
    dec = RangeDecoder(data)
    out: List[int] = []
    for i in range(n):
        freqs = next_freqs(i, out)
        cum = _cumulate(freqs)
        tot = int(cum[-1])
        target = dec.decode_cum(tot)
        s = int(np.searchsorted(cum, target, side="right") - 1)
        dec.decode_update(int(cum[s]), int(freqs[s]), tot)
        out.append(s)
    return out
