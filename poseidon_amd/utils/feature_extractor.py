"""Feature extraction: load a .caffemodel, run forward passes, dump named
blobs to a PDB per rank.

Parity: /root/reference/src/caffe/feature_extractor.cpp:16-135 +
tools/extract_features.cpp (per-(client,thread) LevelDBs become per-rank
PDB files; PS SyncWithPS becomes an RCCL broadcast at load).
"""

from __future__ import annotations

from typing import List, Sequence

import numpy as np
import torch

from ..core.context import ctx
from ..core.net import Net, TEST
from ..data.pdb import PDBWriter, array_to_datum
from ..proto import Message, read_proto_binary, read_proto_text
from ..parallel import comm


class FeatureExtractor:
    def __init__(self, net_param: Message, weights_file: str):
        self.net = Net(net_param, phase=TEST)
        proto = read_proto_binary(weights_file, "NetParameter")
        self.net.copy_trained_layers_from(proto)
        if ctx().distributed and comm.init_distributed():
            comm.broadcast_params(
                [ps for i, ps in enumerate(self.net.params) if ps.owner == i])

    def extract(self, blob_names: Sequence[str], num_batches: int,
                out_prefix: str) -> List[str]:
        c = ctx()
        writers = {name: PDBWriter(f"{out_prefix}_{name}_{c.rank}.pdb")
                   for name in blob_names}
        for _ in range(num_batches):
            self.net.forward()
            for name in blob_names:
                blob = self.net.blobs[name]
                feats = blob.data.detach().to(torch.float32).cpu().numpy()
                feats = feats.reshape(feats.shape[0], -1)
                for row in feats:
                    arr = row.reshape(1, 1, -1).astype(np.float32)
                    writers[name].put(array_to_datum(arr, 0))
        paths = []
        for name, w in writers.items():
            w.close()
            paths.append(w.path)
        return paths


def extract_features_cli(argv=None):
    import argparse
    ap = argparse.ArgumentParser(description="poseidon_amd feature extractor")
    ap.add_argument("--model", required=True, help=".caffemodel weights")
    ap.add_argument("--net", required=True, help="NetParameter prototxt")
    ap.add_argument("--blobs", required=True,
                    help="comma-separated blob names to extract")
    ap.add_argument("--batches", type=int, default=10)
    ap.add_argument("--out", required=True, help="output PDB path prefix")
    args = ap.parse_args(argv)

    import poseidon_amd as pa
    pa.init(device="cuda" if torch.cuda.is_available() else "cpu")
    net_param = read_proto_text(args.net, "NetParameter")
    fx = FeatureExtractor(net_param, args.model)
    paths = fx.extract(args.blobs.split(","), args.batches, args.out)
    print("wrote:", *paths, sep="\n  ")
