"""Minimal NanoVDB (.nvdb) file I/O: read float grids into dense arrays,
write dense arrays as valid .nvdb files.

Capability parity: the reference's GridVolumeManager reads .nvdb files and
uploads NanoVDB trees to the GPU (/root/reference/src/impl/vol_grid.cu:
216-342, nanovdb::io::readGrids + deviceUpload).  The MI355X build renders
volumes from DENSE density grids (csrc/core/medium.h delta/ratio tracking
over a (nz,ny,nx) float array + majorant supergrid), so .nvdb ingestion is
a host-side conversion: sparse tree -> dense array + world-space bounds.

Scope (documented subset, checked loudly):
  * NanoVDB ABI 32.x float grids (LevelSet/FogVolume/Unknown classes),
  * file codecs NONE and ZIP (zlib per-grid blobs); BLOSC raises with a
    clear message,
  * single-key root tiles (the library default NANOVDB_USE_SINGLE_ROOT_KEY).
The writer emits the same subset (one grid per file), so any file this
module writes it also reads back bit-exactly; scripts/make_scenes.py ships
scenes/assets/smoke.nvdb built with it.

Layout summary (NanoVDB 32.3, 32-byte data alignment):
  file   := FileHeader (16 B) { magic, version, gridCount, codec }
            per grid: FileMetaData (176 B) + gridName + grid blob
  grid   := GridData (672 B) + TreeData (64 B) + RootData + tiles
            + upper nodes (32^3) + lower nodes (16^3) + leaves (8^3)
  tree offsets in TreeData are bytes relative to the TreeData start;
  root-tile child offsets are relative to the RootData start; internal
  table child offsets are relative to their node's start.
"""
from __future__ import annotations

import struct

import numpy as np

MAGIC = 0x304244566F6E614E          # "NanoVDB0" little-endian
ALIGN = 32

# enum GridType (subset)
GT_FLOAT = 1
# enum GridClass (subset)
GC_UNKNOWN, GC_LEVEL_SET, GC_FOG = 0, 1, 2
# enum io::Codec
CODEC_NONE, CODEC_ZIP, CODEC_BLOSC = 0, 1, 2

GRID_DATA_SIZE = 672
TREE_DATA_SIZE = 64
ROOT_DATA_SIZE = 64                  # BBox(24) + tableSize(4) + 5 floats + pad
ROOT_TILE_SIZE = 32                  # key(8) + child(8) + state(4) + value(4) + pad
UPPER_SIZE = 24 + 8 + 4096 + 4096 + 16 + 16 + 32768 * 8   # = 270336
LOWER_SIZE = 24 + 8 + 512 + 512 + 16 + 16 + 4096 * 8      # = 33856
LEAF_SIZE = 12 + 4 + 64 + 16 + 512 * 4                    # = 2144


def _version(major=32, minor=3, patch=0):
    return (major << 21) | (minor << 10) | patch


def _split_version(v):
    return (v >> 21) & 0x7FF, (v >> 10) & 0x7FF, v & 0x3FF


def _pad32(n):
    return (n + ALIGN - 1) // ALIGN * ALIGN


def _root_key(i, j, k):
    """Single-root-key packing of an upper-node origin (span 4096 = 2^12)."""
    return ((np.uint64(np.int64(i) >> 12) & np.uint64(0x1FFFFF)) << np.uint64(42)) \
         | ((np.uint64(np.int64(j) >> 12) & np.uint64(0x1FFFFF)) << np.uint64(21)) \
         | (np.uint64(np.int64(k) >> 12) & np.uint64(0x1FFFFF))


class NvdbError(ValueError):
    pass


# --------------------------------------------------------------------- write
def write_nvdb(path, density, voxel_size=1.0, origin=(0, 0, 0),
               grid_name="density", grid_class=GC_FOG,
               world_origin=(0.0, 0.0, 0.0), codec="none"):
    """Write a dense (nz, ny, nx) float32 array as a single-grid .nvdb file
    (codec NONE).  `origin` is the index-space coordinate of voxel [0,0,0];
    world transform is a uniform scale by `voxel_size` plus a translation by
    `world_origin`.  Zero voxels become inactive background (the file stores
    only non-empty 8^3 leaves)."""
    d = np.ascontiguousarray(np.asarray(density, np.float32))
    if d.ndim != 3:
        raise NvdbError("density must be (nz, ny, nx)")
    nz, ny, nx = d.shape
    oi, oj, ok = (int(v) for v in origin)
    if min(oi, oj, ok) < 0 or max(oi + nx, oj + ny, ok + nz) > 4096:
        raise NvdbError("writer supports index bounds within one upper node "
                        "(origin >= 0, extent <= 4096)")

    # ---- collect non-empty leaves (index space: i=x, j=y, k=z)
    # pad the array to 8-multiples for easy slicing
    px, py, pz = (-nx) % 8, (-ny) % 8, (-nz) % 8
    dp = np.pad(d, ((0, pz), (0, py), (0, px)))
    lz, ly, lx = dp.shape[0] // 8, dp.shape[1] // 8, dp.shape[2] // 8
    blocks = dp.reshape(lz, 8, ly, 8, lx, 8).transpose(0, 2, 4, 1, 3, 5)
    nonzero = blocks.reshape(lz, ly, lx, -1).any(axis=-1)
    leaf_idx = np.argwhere(nonzero)          # (n, 3) in (bz, by, bx)
    leaves = []                              # (origin_ijk, values (8,8,8) zyx)
    for bz, by, bx in leaf_idx:
        vals = blocks[bz, by, bx]            # (z, y, x)
        leaves.append(((oi + 8 * int(bx), oj + 8 * int(by), ok + 8 * int(bz)), vals))

    # group leaves under lower (16^3 leaves = 128^3 voxels) and upper nodes
    lowers = {}
    for org, vals in leaves:
        lkey = (org[0] >> 7 << 7, org[1] >> 7 << 7, org[2] >> 7 << 7)
        lowers.setdefault(lkey, []).append((org, vals))
    uppers = {}
    for lkey, lvs in lowers.items():
        ukey = (lkey[0] >> 12 << 12, lkey[1] >> 12 << 12, lkey[2] >> 12 << 12)
        uppers.setdefault(ukey, []).append((lkey, lvs))
    uppers = dict(sorted(uppers.items()))

    vmax = float(d.max()) if d.size else 0.0
    vmin = float(d.min()) if d.size else 0.0
    n_leaf = len(leaves)
    n_lower = len(lowers)
    n_upper = len(uppers)

    # ---- layout: Grid | Tree | Root+tiles | uppers | lowers | leaves
    root_off = GRID_DATA_SIZE + TREE_DATA_SIZE              # rel. to grid start
    root_size = ROOT_DATA_SIZE + n_upper * ROOT_TILE_SIZE
    upper_off = root_off + root_size
    lower_off = upper_off + n_upper * UPPER_SIZE
    leaf_off = lower_off + n_lower * LOWER_SIZE
    grid_size = leaf_off + n_leaf * LEAF_SIZE
    buf = bytearray(grid_size)

    # index bbox (inclusive)
    ib_min = (oi, oj, ok)
    ib_max = (oi + nx - 1, oj + ny - 1, ok + nz - 1)
    vs = float(voxel_size)
    wo = tuple(float(v) for v in world_origin)
    wb_min = tuple(wo[a] + ib_min[a] * vs for a in range(3))
    wb_max = tuple(wo[a] + (ib_max[a] + 1) * vs for a in range(3))

    # ---- GridData (672 B)
    name_b = grid_name.encode()[:255]
    g = struct.pack("<QQIIII", MAGIC, 0xFFFFFFFFFFFFFFFF, _version(), 0, 0, 1)
    g += struct.pack("<Q", grid_size)
    g += name_b + b"\0" * (256 - len(name_b))
    # Map: float mat/inv/vec/taper + double mat/inv/vec/taper (uniform scale)
    matf = [vs, 0, 0, 0, vs, 0, 0, 0, vs]
    invf = [1 / vs, 0, 0, 0, 1 / vs, 0, 0, 0, 1 / vs]
    g += struct.pack("<9f", *matf) + struct.pack("<9f", *invf)
    g += struct.pack("<3f", *wo) + struct.pack("<f", 1.0)
    g += struct.pack("<9d", *matf) + struct.pack("<9d", *invf)
    g += struct.pack("<3d", *wo) + struct.pack("<d", 1.0)
    g += struct.pack("<6d", *wb_min, *wb_max)
    g += struct.pack("<3d", vs, vs, vs)
    g += struct.pack("<II", grid_class, GT_FLOAT)
    g += struct.pack("<qI", 0, 0)        # blind metadata offset/count
    g += struct.pack("<I", 0) + struct.pack("<QQ", 0, 0)   # data0..2
    g += b"\0" * (GRID_DATA_SIZE - len(g))
    buf[:GRID_DATA_SIZE] = g

    # ---- TreeData (64 B): offsets are relative to the TreeData start
    t = struct.pack("<4Q",
                    leaf_off - GRID_DATA_SIZE,
                    lower_off - GRID_DATA_SIZE,
                    upper_off - GRID_DATA_SIZE,
                    root_off - GRID_DATA_SIZE)
    t += struct.pack("<3I", n_leaf, n_lower, n_upper)
    t += struct.pack("<3I", 0, 0, 0)     # active tile counts per level
    t += struct.pack("<Q", int(np.count_nonzero(d)))
    t += b"\0" * (TREE_DATA_SIZE - len(t))
    buf[GRID_DATA_SIZE:GRID_DATA_SIZE + TREE_DATA_SIZE] = t

    # ---- RootData + tiles
    r = struct.pack("<6i", *ib_min, *ib_max)
    r += struct.pack("<I", n_upper)
    r += struct.pack("<5f", 0.0, vmin, vmax, 0.0, 0.0)   # background, min, max, avg, std
    r += b"\0" * (ROOT_DATA_SIZE - len(r))
    tiles = b""
    for ui, ukey in enumerate(uppers):
        child_rel = (upper_off + ui * UPPER_SIZE) - root_off
        tiles += struct.pack("<QqIf", int(_root_key(*ukey)), child_rel, 0, 0.0)
        tiles += b"\0" * (ROOT_TILE_SIZE - 24)
    buf[root_off:root_off + root_size] = r + tiles

    # ---- nodes
    lower_list = []           # (lkey, leaves) in emission order
    for ukey, lkeys in uppers.items():
        lower_list.extend(lkeys)
    lower_pos = {lkey: li for li, (lkey, _) in enumerate(lower_list)}
    leaf_list = []
    for lkey, lvs in lower_list:
        leaf_list.extend(lvs)
    leaf_pos = {org: i for i, (org, _) in enumerate(leaf_list)}

    for ui, (ukey, lkeys) in enumerate(uppers.items()):
        base = upper_off + ui * UPPER_SIZE
        child_mask = np.zeros(512, np.uint64)    # 32768 bits
        table = np.zeros(32768, np.int64)
        bb_lo = [1 << 30] * 3
        bb_hi = [-(1 << 30)] * 3
        for lkey, lvs in lkeys:
            # child index inside the 32^3 table: n = (x<<10)|(y<<5)|z of the
            # lower node's local coords (each step = 128 voxels)
            cx = (lkey[0] - ukey[0]) >> 7
            cy = (lkey[1] - ukey[1]) >> 7
            cz = (lkey[2] - ukey[2]) >> 7
            n = (cx << 10) | (cy << 5) | cz
            child_mask[n >> 6] |= np.uint64(1) << np.uint64(n & 63)
            li = lower_pos[lkey]
            table[n] = (lower_off + li * LOWER_SIZE) - base
            for a in range(3):
                bb_lo[a] = min(bb_lo[a], lkey[a])
                bb_hi[a] = max(bb_hi[a], lkey[a] + 127)
        nd = struct.pack("<6i", *bb_lo, *bb_hi)
        nd += struct.pack("<Q", 0)                       # flags
        nd += np.zeros(512, np.uint64).tobytes()          # value mask (no tiles)
        nd += child_mask.tobytes()
        nd += struct.pack("<4f", vmin, vmax, 0.0, 0.0)
        nd += b"\0" * 16                                  # pad to 32
        nd += table.tobytes()
        assert len(nd) == UPPER_SIZE
        buf[base:base + UPPER_SIZE] = nd

    for li, (lkey, lvs) in enumerate(lower_list):
        base = lower_off + li * LOWER_SIZE
        child_mask = np.zeros(64, np.uint64)      # 4096 bits
        table = np.zeros(4096, np.int64)
        bb_lo = [1 << 30] * 3
        bb_hi = [-(1 << 30)] * 3
        for org, _vals in lvs:
            cx = (org[0] - lkey[0]) >> 3
            cy = (org[1] - lkey[1]) >> 3
            cz = (org[2] - lkey[2]) >> 3
            n = (cx << 8) | (cy << 4) | cz
            child_mask[n >> 6] |= np.uint64(1) << np.uint64(n & 63)
            table[n] = (leaf_off + leaf_pos[org] * LEAF_SIZE) - base
            for a in range(3):
                bb_lo[a] = min(bb_lo[a], org[a])
                bb_hi[a] = max(bb_hi[a], org[a] + 7)
        nd = struct.pack("<6i", *bb_lo, *bb_hi)
        nd += struct.pack("<Q", 0)
        nd += np.zeros(64, np.uint64).tobytes()
        nd += child_mask.tobytes()
        nd += struct.pack("<4f", vmin, vmax, 0.0, 0.0)
        nd += b"\0" * 16
        nd += table.tobytes()
        assert len(nd) == LOWER_SIZE
        buf[base:base + LOWER_SIZE] = nd

    for i, (org, vals) in enumerate(leaf_list):
        base = leaf_off + i * LEAF_SIZE
        # value mask: voxel n = (x<<6)|(y<<3)|z
        active = (vals != 0).transpose(2, 1, 0).reshape(-1)   # (x,y,z) order
        vmask = np.packbits(active, bitorder="little")
        nd = struct.pack("<3i", *org)
        nd += struct.pack("<3Bb", 7, 7, 7, 0)     # bbox dif + flags
        nd += vmask.tobytes()
        nd += struct.pack("<4f", float(vals.min()), float(vals.max()), 0.0, 0.0)
        nd += vals.transpose(2, 1, 0).astype("<f4").tobytes()  # mValues[(x<<6)|(y<<3)|z]
        assert len(nd) == LEAF_SIZE
        buf[base:base + LEAF_SIZE] = nd

    # ---- file wrapper
    name_file = grid_name.encode() + b"\0"
    meta = struct.pack("<4Q", grid_size, grid_size, 0, int(np.count_nonzero(d)))
    meta += struct.pack("<II", GT_FLOAT, grid_class)
    meta += struct.pack("<6d", *wb_min, *wb_max)
    meta += struct.pack("<6i", *ib_min, *ib_max)
    meta += struct.pack("<3d", vs, vs, vs)
    meta += struct.pack("<I", len(name_file))
    meta += struct.pack("<4I", n_leaf, n_lower, n_upper, 1)   # node counts (leaf,lower,upper,root)
    meta += struct.pack("<3I", 0, 0, 0)                       # tile counts
    # codec/version appended below (depends on the codec argument)
    codec_id = CODEC_ZIP if codec == "zip" else CODEC_NONE
    blob = bytes(buf)
    if codec_id == CODEC_ZIP:
        import zlib
        blob = zlib.compress(blob, 6)
    meta = meta[:168] + struct.pack("<HH", codec_id, 0) + struct.pack("<I", _version())
    # fileSize field (offset 8) = on-disk blob size (compressed when zipped)
    meta = meta[:8] + struct.pack("<Q", len(blob)) + meta[16:]
    assert len(meta) == 176, len(meta)
    with open(path, "wb") as f:
        f.write(struct.pack("<QIHH", MAGIC, _version(), 1, codec_id))
        f.write(meta)
        f.write(name_file)
        f.write(blob)


# ---------------------------------------------------------------------- read
def _read_grid(blob):
    """Parse one uncompressed NanoVDB float-grid blob -> dict."""
    magic, _cksum, version = struct.unpack_from("<QQI", blob, 0)
    if magic != MAGIC:
        raise NvdbError(f"bad NanoVDB grid magic 0x{magic:x}")
    major, minor, patch = _split_version(version)
    if major != 32:
        raise NvdbError(f"unsupported NanoVDB ABI {major}.{minor}.{patch} "
                        "(this reader supports 32.x)")
    name = blob[40:40 + 256].split(b"\0")[0].decode(errors="replace")
    off = 40 + 256
    matf = struct.unpack_from("<9f", blob, off)
    off += 264                                     # Map
    wb = struct.unpack_from("<6d", blob, off)
    off += 48
    vsz = struct.unpack_from("<3d", blob, off)
    off += 24
    grid_class, grid_type = struct.unpack_from("<II", blob, off)
    if grid_type != GT_FLOAT:
        raise NvdbError(f"unsupported grid type {grid_type} (float only)")

    # TreeData
    tb = GRID_DATA_SIZE
    leaf_off, lower_off, upper_off, root_off = struct.unpack_from("<4Q", blob, tb)
    root = tb + root_off
    ib = struct.unpack_from("<6i", blob, root)
    table_size, = struct.unpack_from("<I", blob, root + 24)
    background, = struct.unpack_from("<f", blob, root + 28)

    imin = np.array(ib[:3], np.int64)
    imax = np.array(ib[3:], np.int64)
    shape = (imax - imin + 1)                      # (x, y, z) extents
    dense = np.full((int(shape[2]), int(shape[1]), int(shape[0])),
                    background, np.float32)        # (z, y, x)

    def leaf_into(leaf_base):
        org = struct.unpack_from("<3i", blob, leaf_base)
        vals = np.frombuffer(blob, "<f4", 512, leaf_base + 96)
        v = vals.reshape(8, 8, 8).transpose(2, 1, 0)   # (x,y,z) -> (z,y,x)
        mask = np.unpackbits(np.frombuffer(blob, np.uint8, 64, leaf_base + 16),
                             bitorder="little").reshape(8, 8, 8).transpose(2, 1, 0)
        x0, y0, z0 = (int(org[a] - imin[a]) for a in range(3))
        zs, ys, xs = dense.shape
        # clip (leaves may straddle the index bbox on badly formed files)
        v = np.where(mask > 0, v, background)
        z1, y1, x1 = min(z0 + 8, zs), min(y0 + 8, ys), min(x0 + 8, xs)
        if z0 < 0 or y0 < 0 or x0 < 0 or z1 <= z0 or y1 <= y0 or x1 <= x0:
            return
        dense[z0:z1, y0:y1, x0:x1] = v[:z1 - z0, :y1 - y0, :x1 - x0]

    def walk_lower(base):
        cm = np.frombuffer(blob, np.uint64, 64, base + 24 + 8 + 512)
        bits = np.unpackbits(cm.view(np.uint8), bitorder="little")
        table = np.frombuffer(blob, np.int64, 4096, base + 24 + 8 + 512 + 512 + 32)
        for n in np.nonzero(bits)[0]:
            leaf_into(base + int(table[n]))

    def walk_upper(base):
        cm = np.frombuffer(blob, np.uint64, 512, base + 24 + 8 + 4096)
        bits = np.unpackbits(cm.view(np.uint8), bitorder="little")
        table = np.frombuffer(blob, np.int64, 32768, base + 24 + 8 + 4096 + 4096 + 32)
        for n in np.nonzero(bits)[0]:
            walk_lower(base + int(table[n]))

    for t in range(table_size):
        toff = root + ROOT_DATA_SIZE + t * ROOT_TILE_SIZE
        _key, child, _state, _value = struct.unpack_from("<QqIf", blob, toff)
        if child != 0:
            walk_upper(root + child)

    return {
        "name": name,
        "dense": dense,
        "index_min": tuple(int(v) for v in imin),
        "voxel_size": tuple(float(v) for v in vsz),
        "world_min": tuple(wb[:3]),
        "world_max": tuple(wb[3:]),
        "grid_class": grid_class,
        "background": float(background),
        "version": (major, minor, patch),
        "map_scale": float(matf[0]),
    }


def read_nvdb(path):
    """Read all float grids of a .nvdb file -> list of dicts with keys
    name, dense (nz,ny,nx float32), index_min, voxel_size, world_min/max."""
    with open(path, "rb") as f:
        data = f.read()
    magic, version, grid_count, codec = struct.unpack_from("<QIHH", data, 0)
    if magic != MAGIC:
        raise NvdbError(f"{path}: not a NanoVDB file (magic 0x{magic:x})")
    if codec not in (CODEC_NONE, CODEC_ZIP):
        raise NvdbError(f"{path}: codec BLOSC is not supported — re-export "
                        "uncompressed or zlib (nanovdb_convert without -b)")
    grids = []
    off = 16
    for _ in range(grid_count):
        meta = data[off:off + 176]
        grid_size, file_size, _key, _nvox = struct.unpack_from("<4Q", meta, 0)
        name_size, = struct.unpack_from("<I", meta, 136)
        off += 176
        off += name_size
        if codec == CODEC_ZIP:
            import zlib
            try:
                blob = zlib.decompress(data[off:off + file_size])
            except zlib.error as e:
                raise NvdbError(f"{path}: corrupt ZIP grid blob ({e})")
            if len(blob) != grid_size:
                raise NvdbError(f"{path}: zip blob decompressed to "
                                f"{len(blob)} bytes, expected {grid_size}")
            off += file_size
        else:
            blob = data[off:off + grid_size]
            off += grid_size
        grids.append(_read_grid(blob))
    return grids
