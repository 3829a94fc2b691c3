"""Convert DeepMind learning-to-simulate Water-3D TFRecords to HDF5.

Re-owned equivalent of the reference converter
(dataset_generation/Water-3D/tfrecord_to_h5.py): each trajectory becomes an
HDF5 group with ``particle_type`` [N] and ``position`` [T, N, 3], which the
training preprocessing reads (distegnn_amd/data/preprocess.py). Requires
TensorFlow (TFRecord reader) and h5py — run wherever those are available
and copy the .h5 files into ``data.data_dir``.
"""

import argparse
import functools
import json
import os


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset-dir", required=True,
                    help="dir with {train,valid,test}.tfrecord + metadata.json")
    ap.add_argument("--out-dir", required=True)
    args = ap.parse_args()
    try:
        import h5py
        import numpy as np
        import tensorflow as tf
    except ImportError as e:
        raise SystemExit(f"this converter needs TensorFlow + h5py: {e}")

    with open(os.path.join(args.dataset_dir, "metadata.json")) as f:
        meta = json.load(f)
    dim = meta["dim"]
    os.makedirs(args.out_dir, exist_ok=True)

    def parse(proto):
        ctx, feats = tf.io.parse_single_sequence_example(
            proto,
            context_features={
                "particle_type": tf.io.VarLenFeature(tf.string),
                "key": tf.io.FixedLenFeature([], tf.int64),
            },
            sequence_features={
                "position": tf.io.VarLenFeature(tf.string)})
        ptype = tf.io.decode_raw(tf.sparse.to_dense(
            ctx["particle_type"])[0], tf.int64)
        pos = tf.map_fn(
            functools.partial(tf.io.decode_raw, out_type=tf.float32),
            tf.sparse.to_dense(feats["position"]), fn_output_signature=tf.float32)
        return ptype, pos

    for split in ("train", "valid", "test"):
        path = os.path.join(args.dataset_dir, f"{split}.tfrecord")
        if not os.path.exists(path):
            print(f"skip {split} (no {path})")
            continue
        ds = tf.data.TFRecordDataset([path]).map(parse)
        out = os.path.join(args.out_dir, f"{split}.h5")
        with h5py.File(out, "w") as h5:
            for i, (ptype, pos) in enumerate(ds):
                g = h5.create_group(str(i))
                p = pos.numpy().reshape(pos.shape[0], -1, dim)
                g.create_dataset("particle_type", data=ptype.numpy())
                g.create_dataset("position", data=p)
        print(f"{split} -> {out}")


if __name__ == "__main__":
    main()
