"""The shipped models/*.prototxt and examples/*_solver.prototxt must parse,
round-trip through the text codec, and build (guards the generated
artifacts against codec or zoo drift)."""

import glob
import os

import poseidon_amd as pa
from poseidon_amd.core.net import Net, TRAIN
from poseidon_amd.proto import parse_text, read_proto_text, to_text

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_model_prototxts_parse_and_build():
    paths = sorted(glob.glob(os.path.join(REPO, "models", "*.prototxt")))
    assert len(paths) >= 5
    pa.init(device="cpu", seed=1)
    for p in paths:
        np_ = read_proto_text(p, "NetParameter")
        # text round-trip is stable
        again = parse_text("NetParameter", to_text(np_))
        assert again.encode() == np_.encode(), p
        # shrink the batch so building is fast
        for lp in np_.layers:
            dp = lp.dummy_data_param
            if dp is not None and len(list(dp.num)):
                for i in range(len(dp.num)):
                    dp.num[i] = 2
        net = Net(np_, phase=TRAIN, verbose=False)
        assert net.layers, p


def test_example_solvers_parse():
    paths = sorted(glob.glob(os.path.join(REPO, "examples", "*.prototxt")))
    assert len(paths) >= 3
    for p in paths:
        sp = read_proto_text(p, "SolverParameter")
        assert sp.base_lr > 0, p
        assert sp.has("net"), p
