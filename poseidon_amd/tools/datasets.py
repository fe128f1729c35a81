"""Dataset tooling: image list -> PDB, mean computation, round-robin
partitioning, device query.

Parity: reference tools/{convert_imageset,compute_image_mean,
partition_data}.cpp and caffe_main device_query (LMDB/LevelDB replaced by
the PDB container; OpenCV by PIL).
"""

from __future__ import annotations

import argparse

import numpy as np

from ..data.pdb import PDBReader, PDBWriter, array_to_datum, datum_to_array
from ..proto import Message, write_proto_binary


def convert_imageset(argv=None):
    """Image list file ("path label" lines) -> PDB of encoded Datums."""
    ap = argparse.ArgumentParser()
    ap.add_argument("listfile", help='lines of "image_path label"')
    ap.add_argument("out_pdb")
    ap.add_argument("--resize", type=int, default=0,
                    help="resize to NxN before storing")
    ap.add_argument("--shuffle", action="store_true")
    args = ap.parse_args(argv)
    from PIL import Image

    with open(args.listfile) as f:
        entries = [ln.split() for ln in f if ln.strip()]
    if args.shuffle:
        np.random.default_rng(0).shuffle(entries)
    with PDBWriter(args.out_pdb) as w:
        for path, label in entries:
            img = Image.open(path).convert("RGB")
            if args.resize:
                img = img.resize((args.resize, args.resize))
            arr = np.asarray(img, dtype=np.uint8)[:, :, ::-1]  # BGR like Caffe
            chw = np.ascontiguousarray(arr.transpose(2, 0, 1))
            w.put(array_to_datum(chw, int(label)))
    print(f"wrote {len(entries)} records -> {args.out_pdb}")


def compute_image_mean(argv=None):
    """PDB -> mean BlobProto (reference compute_image_mean.cpp)."""
    ap = argparse.ArgumentParser()
    ap.add_argument("pdb")
    ap.add_argument("out_blobproto")
    args = ap.parse_args(argv)
    db = PDBReader(args.pdb)
    acc = None
    for datum in db:
        arr = datum_to_array(datum)
        acc = arr.astype(np.float64) if acc is None else acc + arr
    mean = (acc / len(db)).astype(np.float32)
    proto = Message("BlobProto", num=1, channels=mean.shape[0],
                    height=mean.shape[1], width=mean.shape[2])
    proto.data = mean.ravel()
    write_proto_binary(proto, args.out_blobproto)
    print(f"mean over {len(db)} images -> {args.out_blobproto}")


def partition_data(argv=None):
    """Round-robin split a PDB into per-rank shards source_0..source_{n-1}
    (reference partition_data.cpp:27-148)."""
    ap = argparse.ArgumentParser()
    ap.add_argument("pdb")
    ap.add_argument("num_shards", type=int)
    args = ap.parse_args(argv)
    db = PDBReader(args.pdb)
    writers = [PDBWriter(f"{args.pdb}_{i}") for i in range(args.num_shards)]
    for i in range(len(db)):
        writers[i % args.num_shards].put_raw(db.get_raw(i))
    for w in writers:
        w.close()
    print(f"{len(db)} records -> {args.num_shards} shards")


def convert_db(argv=None):
    """Convert between the PDB container and LMDB environments in either
    direction -- lets reference-produced LMDB shards run here unmodified
    and lets PDB datasets be exported for the reference's own tools."""
    ap = argparse.ArgumentParser()
    ap.add_argument("src", help="PDB file or LMDB env dir / data.mdb")
    ap.add_argument("dst", help="output PDB file (or LMDB dir with --to-lmdb)")
    ap.add_argument("--to-lmdb", action="store_true")
    args = ap.parse_args(argv)
    from ..data.lmdb_io import LmdbReader, LmdbWriter
    from ..data.leveldb_io import LevelDbReader
    import os

    def open_src(p):
        if os.path.isdir(p):
            if os.path.exists(os.path.join(p, "data.mdb")):
                return LmdbReader(p)
            return LevelDbReader(p)  # reference create_*.sh default
        if p.endswith(".mdb"):
            return LmdbReader(p)
        return PDBReader(p)

    db = open_src(args.src)
    if args.to_lmdb:
        with LmdbWriter(args.dst) as w:
            for i in range(len(db)):
                w.put(b"%08d" % i, db.get_raw(i))
    else:
        with PDBWriter(args.dst) as w:
            for i in range(len(db)):
                w.put_raw(db.get_raw(i))
    print(f"converted {len(db)} records {args.src} -> {args.dst}")


def device_query(argv=None):
    """caffe_main device_query equivalent."""
    import torch
    if not torch.cuda.is_available():
        print("no GPU visible")
        return
    for i in range(torch.cuda.device_count()):
        p = torch.cuda.get_device_properties(i)
        print(f"GPU {i}: {p.name} gcnArch={getattr(p, 'gcnArchName', '?')} "
              f"CUs={p.multi_processor_count} "
              f"HBM={p.total_memory / (1 << 30):.0f}GB")


if __name__ == "__main__":
    import sys
    cmds = {"convert_imageset": convert_imageset,
            "compute_image_mean": compute_image_mean,
            "partition_data": partition_data,
            "convert_db": convert_db,
            "device_query": device_query}
    if len(sys.argv) < 2 or sys.argv[1] not in cmds:
        print("usage: python -m poseidon_amd.tools.datasets "
              f"{{{','.join(cmds)}}} ...")
        raise SystemExit(2)
    cmds[sys.argv[1]](sys.argv[2:])
