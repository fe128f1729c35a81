"""Tools + feature extractor + dataset pipeline tests (CPU)."""

import os

import numpy as np

import poseidon_amd as pa
from poseidon_amd.data.pdb import PDBReader, PDBWriter, array_to_datum, datum_to_array
from poseidon_amd.proto import Message, parse_text, to_text, write_proto_text
from poseidon_amd.models import zoo


def _make_pdb(path, n=20, c=3, hw=8, classes=4, seed=0):
    rng = np.random.default_rng(seed)
    with PDBWriter(path) as w:
        for i in range(n):
            arr = rng.integers(0, 255, (c, hw, hw), dtype=np.uint8)
            w.put(array_to_datum(arr.astype(np.uint8), int(i % classes)))


def test_pdb_roundtrip(tmp_path):
    p = str(tmp_path / "toy.pdb")
    _make_pdb(p)
    db = PDBReader(p)
    assert len(db) == 20
    d = db.get(3)
    arr = datum_to_array(d)
    assert arr.shape == (3, 8, 8)
    assert d.label == 3


def test_partition_and_mean(tmp_path):
    from poseidon_amd.tools.datasets import partition_data, compute_image_mean
    p = str(tmp_path / "toy.pdb")
    _make_pdb(p)
    partition_data([p, "3"])
    sizes = [len(PDBReader(f"{p}_{i}")) for i in range(3)]
    assert sum(sizes) == 20 and max(sizes) - min(sizes) <= 1
    mean_path = str(tmp_path / "mean.binaryproto")
    compute_image_mean([p, mean_path])
    from poseidon_amd.proto import read_proto_binary
    proto = read_proto_binary(mean_path, "BlobProto")
    assert proto.channels == 3 and proto.height == 8


def test_data_layer_from_pdb(tmp_path):
    pa.init(device="cpu", seed=1)
    p = str(tmp_path / "train.pdb")
    _make_pdb(p, n=32)
    net_param = parse_text("NetParameter", f"""
        name: "pdbnet"
        layers {{ name: "data" type: DATA top: "data" top: "label"
                 data_param {{ source: "{p}" batch_size: 8 }}
                 transform_param {{ scale: 0.0039 }} }}
        layers {{ name: "ip" type: INNER_PRODUCT bottom: "data" top: "ip"
                 inner_product_param {{ num_output: 4
                    weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "loss" type: SOFTMAX_LOSS bottom: "ip" bottom: "label"
                 top: "loss" }}
    """)
    from poseidon_amd.core.net import Net, TRAIN
    net = Net(net_param, phase=TRAIN)
    loss = net.forward()
    assert np.isfinite(loss)
    assert net.blobs["data"].shape == (8, 3, 8, 8)
    net.backward()


def test_train_cli_end_to_end(tmp_path):
    from poseidon_amd.tools.train import main as train_main
    os.environ.pop("RANK", None)
    os.environ.pop("WORLD_SIZE", None)
    net = zoo.lenet(batch=8)
    net_path = str(tmp_path / "net.prototxt")
    with open(net_path, "w") as f:
        f.write(to_text(net) + "\n")
    solver = Message("SolverParameter", net=net_path, base_lr=0.01,
                     lr_policy="fixed", momentum=0.9, max_iter=3,
                     snapshot_prefix=str(tmp_path / "lenet"),
                     solver_mode="CPU", display=1, random_seed=7)
    sp_path = str(tmp_path / "solver.prototxt")
    write_proto_text(solver, sp_path)
    train_main(["--solver", sp_path, "--net_outputs", str(tmp_path / "out"),
                "--cpu"])
    assert os.path.exists(str(tmp_path / "lenet") + "_iter_3.caffemodel")
    assert os.path.exists(str(tmp_path / "out") + ".netoutputs")
    # finetune path
    train_main(["--solver", sp_path, "--cpu",
                "--weights", str(tmp_path / "lenet") + "_iter_3.caffemodel"])


def test_feature_extractor(tmp_path):
    from poseidon_amd.utils.feature_extractor import FeatureExtractor
    pa.init(device="cpu", seed=2)
    from poseidon_amd.core.net import Net, TRAIN
    from poseidon_amd.proto import write_proto_binary
    net_param = zoo.lenet(batch=4)
    net = Net(net_param, phase=TRAIN)
    model = str(tmp_path / "w.caffemodel")
    write_proto_binary(net.to_proto(), model)

    fx = FeatureExtractor(net_param, model)
    paths = fx.extract(["ip1"], num_batches=2, out_prefix=str(tmp_path / "fx"))
    db = PDBReader(paths[0])
    assert len(db) == 8  # 2 batches x 4
    feat = datum_to_array(db.get(0))
    assert feat.size == 500


def test_dump_prototxt_roundtrip(tmp_path):
    from poseidon_amd.models.zoo import dump_prototxt
    path = str(tmp_path / "alexnet.prototxt")
    dump_prototxt("alexnet", path, batch=4)
    from poseidon_amd.proto import read_proto_text
    net = read_proto_text(path, "NetParameter")
    assert net.name == "AlexNet"
    from poseidon_amd.core.net import Net, TRAIN
    n = Net(net, phase=TRAIN)
    assert any(l.name == "fc6" for l in n.layers)


def test_profile_summary_tool(tmp_path):
    """profile_summary parses a rocpd-schema sqlite DB."""
    import sqlite3
    db = tmp_path / "x_results.db"
    con = sqlite3.connect(str(db))
    con.execute("create table rocpd_kernel_dispatch_ab (id int, kernel_id int,"
                " start int, [end] int)")
    con.execute("create table rocpd_info_kernel_symbol_ab (id int,"
                " display_name text)")
    con.execute("insert into rocpd_info_kernel_symbol_ab values (1, 'k1'),"
                " (2, 'k2')")
    for i in range(4):
        con.execute("insert into rocpd_kernel_dispatch_ab values (?, ?, 0, ?)",
                    (i, 1 + i % 2, 1000 * (i + 1)))
    con.commit()
    from poseidon_amd.tools.profile_summary import summarize
    rows, total = summarize(str(db))
    assert len(rows) == 2
    assert abs(total - 0.01) < 1e-9  # 10,000 ns = 0.01 ms
