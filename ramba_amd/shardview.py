"""View/partition algebra — the MI355X-first redesign of the reference's
`ramba/shardview_array.py` (whole file, 1158 LoC).

The reference stores, per worker, a `[size / index_start / axis_map / steps /
base_offset]` int table (shardview_array.py:32-70) and derives everything
(mapslice :474, mapsv :496, intersect :530, as_base :293, broadcast :978,
remap_axis :1024) from per-worker tables.  Here a view is instead ONE global
affine map from view index space to base global coordinates:

    base_coord[b] = offset[b] + idx[v] * steps[v]      where axis_map[v] == b
    base_coord[b] = offset[b]                          for unmapped base axes

with axis_map[v] == -1 for axes that touch no base axis (np.newaxis-style
broadcast) and steps[v] == 0 for broadcast over an existing base axis.  The
per-worker tables of the reference are recovered by composing this map with the
backing array's division boxes (preimage_box below) — O(1) state per view, and
exactly the (pointer, offset, stride) form a HIP kernel consumes.

Boxes are int64 arrays of shape (2, nd): [0]=lo, [1]=hi INCLUSIVE (the
reference's divisions convention, common.py:309).  A box is empty iff any
hi < lo.
"""

from dataclasses import dataclass
import numpy as np


# ---------------------------------------------------------------------------
# boxes
# ---------------------------------------------------------------------------

def box(lo, hi):
    return np.array([lo, hi], dtype=np.int64)


def box_empty(b):
    return b is None or bool(np.any(b[1] < b[0]))


def box_intersect(a, b):
    if a is None or b is None:
        return None
    lo = np.maximum(a[0], b[0])
    hi = np.minimum(a[1], b[1])
    if np.any(hi < lo):
        return None
    return np.array([lo, hi])


def box_size(b):
    if b is None:
        return 0
    return int(np.prod(b[1] - b[0] + 1))


def box_shape(b):
    return tuple(int(x) for x in (b[1] - b[0] + 1))


def box_contains(outer, inner):
    """True if inner (non-empty) is fully inside outer."""
    return bool(np.all(outer[0] <= inner[0]) and np.all(inner[1] <= outer[1]))


def box_eq(a, b):
    if a is None or b is None:
        return (a is None) == (b is None)
    return bool(np.array_equal(a, b))


def box_subtract(a, b):
    """a minus b as a list of disjoint boxes (standard per-axis slab peel)."""
    if a is None:
        return []
    inter = box_intersect(a, b)
    if inter is None:
        return [a.copy()]
    out = []
    cur = a.copy()
    nd = a.shape[1]
    for d in range(nd):
        if cur[0, d] < inter[0, d]:
            piece = cur.copy()
            piece[1, d] = inter[0, d] - 1
            out.append(piece)
            cur[0, d] = inter[0, d]
        if cur[1, d] > inter[1, d]:
            piece = cur.copy()
            piece[0, d] = inter[1, d] + 1
            out.append(piece)
            cur[1, d] = inter[1, d]
    return out


# ---------------------------------------------------------------------------
# View
# ---------------------------------------------------------------------------

_identity_cache = {}


@dataclass(frozen=True, eq=False)
class View:
    shape: tuple      # view shape
    axis_map: tuple   # len(shape); base axis index, or -1 for a new axis
    steps: tuple      # len(shape); 0 = broadcast over the mapped base axis
    offset: tuple     # len = base ndim; base coord of view index 0 / fixed coord

    def __eq__(self, other):
        # identity fast path: the getitem/identity caches intern Views, so
        # repeated steps compare the same objects
        if self is other:
            return True
        if other.__class__ is not View:
            return NotImplemented
        return (self.shape == other.shape
                and self.axis_map == other.axis_map
                and self.steps == other.steps
                and self.offset == other.offset)

    def __hash__(self):
        # memoised: Views key the hot frontend caches (getitem, arr_var,
        # exec-boxes) and the dataclass hash re-hashes 4 tuples per call
        h = self.__dict__.get("_h")
        if h is None:
            h = hash((self.shape, self.axis_map, self.steps, self.offset))
            object.__setattr__(self, "_h", h)
        return h

    @property
    def ndim(self):
        return len(self.shape)

    @property
    def base_ndim(self):
        return len(self.offset)

    @staticmethod
    def identity(shape):
        shape = tuple(int(s) for s in shape)
        v = _identity_cache.get(shape)
        if v is None:
            nd = len(shape)
            v = View(shape, tuple(range(nd)), (1,) * nd, (0,) * nd)
            if len(_identity_cache) > 4096:
                _identity_cache.clear()
            _identity_cache[shape] = v
        return v

    def is_identity_for(self, base_shape):
        return (self.shape == tuple(base_shape)
                and self.axis_map == tuple(range(len(base_shape)))
                and all(s == 1 for s in self.steps)
                and all(o == 0 for o in self.offset))

    # -- composition --------------------------------------------------------

    def apply_index(self, index):
        """Compose with a basic index (ints / slices / Ellipsis).

        Mirrors the slice branch of getitem_array_executor
        (ramba/ramba.py:6548-6579: slice_distribution + remap_axis) in the
        affine-map representation.
        """
        if not isinstance(index, tuple):
            index = (index,)
        # expand Ellipsis (None/np.newaxis consumes no view axis)
        n_consume = sum(1 for ix in index
                        if ix is not Ellipsis and ix is not None)
        if any(ix is Ellipsis for ix in index):
            k = index.index(Ellipsis)
            fill = (slice(None),) * (self.ndim - n_consume)
            index = index[:k] + fill + index[k + 1:]
            n_consume = self.ndim
        if n_consume > self.ndim:
            raise IndexError(
                f"too many indices for array: array is {self.ndim}-dimensional,"
                f" but {n_consume} were indexed")
        index = index + (slice(None),) * (self.ndim - n_consume)

        new_shape, new_map, new_steps = [], [], []
        offset = list(self.offset)
        v = -1
        for ix in index:
            if ix is None:   # np.newaxis: insert a broadcast axis
                new_shape.append(1)
                new_map.append(-1)
                new_steps.append(0)
                continue
            v += 1
            b, st = self.axis_map[v], self.steps[v]
            if isinstance(ix, (int, np.integer)):
                i = int(ix)
                if i < 0:
                    i += self.shape[v]
                if not (0 <= i < self.shape[v]):
                    raise IndexError(
                        f"index {ix} out of bounds for axis {v} with size"
                        f" {self.shape[v]}")
                if b >= 0:
                    offset[b] += i * st
                # drop the axis
            elif isinstance(ix, slice):
                start, stop, step = ix.indices(self.shape[v])
                length = max(0, -(-(stop - start) // step)) if step > 0 else \
                    max(0, -(-(start - stop) // (-step)))
                new_shape.append(length)
                if b >= 0:
                    offset_delta = start * st
                    offset[b] += offset_delta
                    new_map.append(b)
                    new_steps.append(st * step)
                else:
                    new_map.append(-1)
                    new_steps.append(0)
            else:
                raise IndexError(f"unsupported index element {ix!r} "
                                 "(advanced indexing is out of scope)")
        return View(tuple(new_shape), tuple(new_map), tuple(new_steps),
                    tuple(offset))

    def broadcast_to(self, new_shape):
        """reference: shardview_array.py:978 broadcast."""
        new_shape = tuple(int(s) for s in new_shape)
        nd_new, nd_old = len(new_shape), self.ndim
        if nd_new < nd_old:
            raise ValueError(f"cannot broadcast {self.shape} to {new_shape}")
        lead = nd_new - nd_old
        new_map, new_steps = [], []
        for j in range(nd_new):
            if j < lead:
                new_map.append(-1)
                new_steps.append(0)
                continue
            v = j - lead
            if self.shape[v] == new_shape[j]:
                new_map.append(self.axis_map[v])
                new_steps.append(self.steps[v])
            elif self.shape[v] == 1:
                # pin at the single element's coordinate
                new_map.append(self.axis_map[v])
                new_steps.append(0)
            else:
                raise ValueError(
                    f"cannot broadcast {self.shape} to {new_shape}")
        return View(new_shape, tuple(new_map), tuple(new_steps), self.offset)

    def transpose(self, axes=None):
        """reference: remap_axis, shardview_array.py:1024."""
        if axes is None:
            axes = tuple(reversed(range(self.ndim)))
        assert sorted(axes) == list(range(self.ndim))
        return View(tuple(self.shape[a] for a in axes),
                    tuple(self.axis_map[a] for a in axes),
                    tuple(self.steps[a] for a in axes),
                    self.offset)

    # -- geometry -----------------------------------------------------------

    def full_box(self):
        if any(s == 0 for s in self.shape):
            return None
        lo = np.zeros(self.ndim, dtype=np.int64)
        hi = np.array(self.shape, dtype=np.int64) - 1
        return np.array([lo, hi])

    def image_box(self, vbox):
        """Dense base-space cover of the image of view box `vbox`.

        reference analog: as_base (shardview_array.py:293).
        """
        if box_empty(vbox):
            return None
        lo = np.array(self.offset, dtype=np.int64)
        hi = np.array(self.offset, dtype=np.int64)
        for v in range(self.ndim):
            b, st = self.axis_map[v], self.steps[v]
            if b >= 0 and st != 0:
                a = int(vbox[0, v]) * st
                c = int(vbox[1, v]) * st
                lo[b] += min(a, c)
                hi[b] += max(a, c)
        return np.array([lo, hi])

    def preimage_box(self, bbox):
        """View indices whose image lies inside base box `bbox` (None if empty).

        reference analog: intersect + mapsv (shardview_array.py:530/:496) —
        what slices of the view land on the worker owning `bbox`.
        """
        if bbox is None:
            return None
        lo = np.zeros(self.ndim, dtype=np.int64)
        hi = np.array(self.shape, dtype=np.int64) - 1
        if np.any(hi < lo):
            return None
        mapped = set()
        for v in range(self.ndim):
            b, st = self.axis_map[v], self.steps[v]
            if b < 0:
                continue
            mapped.add(b)
            off = self.offset[b]
            blo, bhi = int(bbox[0, b]), int(bbox[1, b])
            if st == 0:
                if not (blo <= off <= bhi):
                    return None
            elif st > 0:
                lo[v] = max(lo[v], -(-(blo - off) // st))       # ceil
                hi[v] = min(hi[v], (bhi - off) // st)           # floor
            else:
                lo[v] = max(lo[v], -(-(off - bhi) // (-st)))
                hi[v] = min(hi[v], (off - blo) // (-st))
            if hi[v] < lo[v]:
                return None
        for b in range(self.base_ndim):
            if b not in mapped:
                if not (int(bbox[0, b]) <= self.offset[b] <= int(bbox[1, b])):
                    return None
        return np.array([lo, hi])

    # -- kernel addressing --------------------------------------------------

    def operand_addressing(self, i0, cstrides, div_start, border):
        """(offset0, strides) for a kernel operand.

        The kernel iterates local index j over its iteration box; the element
        address (in elements, into the rank's container) is
            offset0 + sum_v j[v] * strides[v].
        `i0` = global view index of the iteration box origin, `cstrides` =
        row-major strides of the container, `div_start` = rank's division box
        start, `border` = border ring width per base axis.
        """
        coord = list(self.offset)
        for v in range(self.ndim):
            b, st = self.axis_map[v], self.steps[v]
            if b >= 0 and st != 0:
                coord[b] += int(i0[v]) * st
        off0 = 0
        for b in range(self.base_ndim):
            off0 += (coord[b] - int(div_start[b]) + int(border[b])) * int(cstrides[b])
        strides = []
        for v in range(self.ndim):
            b, st = self.axis_map[v], self.steps[v]
            strides.append(st * int(cstrides[b]) if b >= 0 else 0)
        return off0, tuple(strides)


def exec_boxes(view, divisions):
    """Per-rank iteration boxes: preimage of each division box under `view`.

    The analog of the reference's exec_dist (`run_deferred_ops` subspace,
    ramba/ramba.py:3497-3500).
    """
    out = []
    for r in range(divisions.shape[0]):
        d = divisions[r]
        if np.any(d[1] < d[0]):
            out.append(None)
            continue
        out.append(view.preimage_box(d))
    return out


def exec_boxes_eq(a, b):
    return len(a) == len(b) and all(box_eq(x, y) for x, y in zip(a, b))
