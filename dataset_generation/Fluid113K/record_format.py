"""Fluid113K record format: write/read helpers + validation.

Each simulation is stored as 16 chunk files
``sim_{idx:04d}_{chunk:02d}.msgpack.zst``; a chunk is a msgpack list of
frames, each ``{"pos": [N,3] float list, "vel": [N,3], "viscosity": [N],
"m": [N]}`` (viscosity/m constant across frames). Requires ``zstandard``.
"""

import argparse


def write_sim(out_dir, idx, frames, viscosity, mass):
    import os

    import msgpack
    import numpy as np
    import zstandard as zstd

    per_chunk = (len(frames) + 15) // 16
    comp = zstd.ZstdCompressor()
    for c in range(16):
        chunk = []
        for fr in frames[c * per_chunk:(c + 1) * per_chunk]:
            chunk.append({"pos": np.asarray(fr[0]).tolist(),
                          "vel": np.asarray(fr[1]).tolist(),
                          "viscosity": np.asarray(viscosity).tolist(),
                          "m": np.asarray(mass).tolist()})
        path = os.path.join(out_dir, f"sim_{idx:04d}_{c:02d}.msgpack.zst")
        with open(path, "wb") as f:
            f.write(comp.compress(msgpack.packb(chunk)))


def validate(path):
    import msgpack
    import zstandard as zstd

    dec = zstd.ZstdDecompressor()
    with open(path, "rb") as f:
        frames = msgpack.unpackb(dec.decompress(f.read()), raw=False)
    assert isinstance(frames, list) and frames, "empty chunk"
    f0 = frames[0]
    for key in ("pos", "vel", "viscosity", "m"):
        assert key in f0, f"missing {key}"
    n = len(f0["pos"])
    assert len(f0["vel"]) == n
    print(f"{path}: {len(frames)} frames, {n} particles — OK")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("path", help="a sim_XXXX_YY.msgpack.zst chunk to validate")
    a = ap.parse_args()
    validate(a.path)
