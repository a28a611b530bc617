"""z-y-x coordinates, bounding boxes and grid decomposition (host side).

Mirrors the behavior of the reference's chunkflow/lib/cartesian_coordinate.py
(Cartesian :34-188, BoundingBox :191-520, BoundingBoxes.from_manual_setup
:524-654) with the same zyx convention and the same clamped-grid task
decomposition; written fresh for this framework (pure Python — the grid math
is negligible next to the HIP hot path it feeds).
"""
from __future__ import annotations

import itertools
from collections import namedtuple
from typing import Union


class Cartesian(namedtuple('Cartesian', ['z', 'y', 'x'])):
    """3D vector in (z, y, x) order with elementwise arithmetic."""
    __slots__ = ()

    @classmethod
    def from_collection(cls, col) -> 'Cartesian':
        if col is None:
            return None
        if isinstance(col, Cartesian):
            return col
        col = tuple(int(v) for v in col)
        assert len(col) == 3
        return cls(*col)

    def _bin(self, other, op):
        if isinstance(other, (tuple, list, Cartesian)):
            return Cartesian(*(op(a, b) for a, b in zip(self, other)))
        return Cartesian(*(op(a, other) for a in self))

    def __add__(self, o): return self._bin(o, lambda a, b: a + b)
    def __radd__(self, o): return self.__add__(o)
    def __sub__(self, o): return self._bin(o, lambda a, b: a - b)
    def __mul__(self, o): return self._bin(o, lambda a, b: a * b)
    def __rmul__(self, o): return self.__mul__(o)
    def __floordiv__(self, o): return self._bin(o, lambda a, b: a // b)
    def __mod__(self, o): return self._bin(o, lambda a, b: a % b)
    def __neg__(self): return Cartesian(-self.z, -self.y, -self.x)

    # elementwise comparisons (all-of semantics, like the reference)
    def __le__(self, o): return all(a <= b for a, b in zip(self, o))
    def __lt__(self, o): return all(a < b for a, b in zip(self, o))
    def __ge__(self, o): return all(a >= b for a, b in zip(self, o))
    def __gt__(self, o): return all(a > b for a, b in zip(self, o))

    def ceildiv(self, o) -> 'Cartesian':
        return Cartesian(*(-(-a // b) for a, b in zip(self, o)))


def to_cartesian(value) -> Union[Cartesian, None]:
    if value is None:
        return None
    return Cartesian.from_collection(value)


class BoundingBox:
    """Axis-aligned box [start, stop) in zyx voxel coordinates."""

    def __init__(self, start, stop):
        self.start = Cartesian.from_collection(start)
        self.stop = Cartesian.from_collection(stop)
        assert self.stop >= self.start

    @classmethod
    def from_delta(cls, start, size) -> 'BoundingBox':
        start = Cartesian.from_collection(start)
        return cls(start, start + Cartesian.from_collection(size))

    @classmethod
    def from_string(cls, s: str) -> 'BoundingBox':
        # "z0-z1_y0-y1_x0-x1"
        parts = s.strip().split('_')
        assert len(parts) == 3
        lo, hi = zip(*(p.split('-') for p in parts))
        return cls(tuple(int(v) for v in lo), tuple(int(v) for v in hi))

    @property
    def string(self) -> str:
        return '_'.join(f'{a}-{b}' for a, b in zip(self.start, self.stop))

    @property
    def shape(self) -> Cartesian:
        return self.stop - self.start

    @property
    def slices(self) -> tuple:
        return tuple(slice(a, b) for a, b in zip(self.start, self.stop))

    def adjust(self, margin) -> 'BoundingBox':
        """Grow (positive margin) or shrink the box symmetrically; margin
        may be a scalar or a zyx triple."""
        if isinstance(margin, int):
            m = Cartesian(margin, margin, margin)
        elif isinstance(margin, Cartesian):
            m = margin
        else:
            m = Cartesian.from_collection(margin[:3])
        return BoundingBox(self.start - m, self.stop + m)

    def clone(self) -> 'BoundingBox':
        return BoundingBox(self.start, self.stop)

    def clamp(self, other: 'BoundingBox') -> 'BoundingBox':
        return BoundingBox(
            Cartesian(*(max(a, b) for a, b in zip(self.start, other.start))),
            Cartesian(*(min(a, b) for a, b in zip(self.stop, other.stop))))

    def contains(self, other: 'BoundingBox') -> bool:
        return self.start <= other.start and other.stop <= self.stop

    def __eq__(self, other):
        return self.start == other.start and self.stop == other.stop

    def __repr__(self):
        return f'BoundingBox({self.string})'


class BoundingBoxes(list):
    """Task decomposition: a grid of chunk bounding boxes over a ROI.

    Matches the reference decomposition
    (lib/cartesian_coordinate.py:598-654): stride = chunk_size - overlap,
    grid = ceil((roi_size - overlap) / stride), box g starts at
    roi_start + g * stride with the full chunk_size shape
    (respect_chunk_size=True) or clipped at roi_stop.
    """

    @classmethod
    def from_manual_setup(cls, chunk_size, chunk_overlap=Cartesian(0, 0, 0),
                          roi_start=None, roi_stop=None, roi_size=None,
                          grid_size=None, respect_chunk_size=True):
        chunk_size = Cartesian.from_collection(chunk_size)
        chunk_overlap = Cartesian.from_collection(chunk_overlap)
        if roi_start is None:
            roi_start = Cartesian(0, 0, 0)
        roi_start = Cartesian.from_collection(roi_start)
        if grid_size is None and roi_size is None and roi_stop is None:
            grid_size = Cartesian(1, 1, 1)
        if roi_size is not None:
            roi_size = Cartesian.from_collection(roi_size)
        elif roi_stop is not None:
            roi_size = Cartesian.from_collection(roi_stop) - roi_start
        stride = chunk_size - chunk_overlap
        if roi_size is None and grid_size is not None:
            grid_size = Cartesian.from_collection(grid_size)
            roi_size = stride * grid_size + chunk_overlap
        if roi_stop is None:
            roi_stop = roi_start + roi_size
        roi_stop = Cartesian.from_collection(roi_stop)
        if grid_size is None:
            grid_size = (roi_size - chunk_overlap).ceildiv(stride)
        for g, s in zip(grid_size, stride):
            if g > 1:
                assert s > 0

        boxes = cls()
        for gz, gy, gx in itertools.product(
                range(grid_size.z), range(grid_size.y), range(grid_size.x)):
            start = roi_start + Cartesian(gz, gy, gx) * stride
            bbox = BoundingBox.from_delta(start, chunk_size)
            if not respect_chunk_size:
                bbox = BoundingBox(
                    bbox.start,
                    Cartesian(*(min(a, b)
                                for a, b in zip(bbox.stop, roi_stop))))
            boxes.append(bbox)
        return boxes
