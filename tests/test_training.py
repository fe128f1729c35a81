"""End-to-end training: solver loop, lr policies, snapshot/restore, and a
convergence smoke test (the reference's acceptance style, SURVEY.md §4:
example configs must reach expected accuracy; here synthetic separable data
replaces MNIST since the image has no datasets)."""

import os

import pytest
import torch

import poseidon_amd as pa


def _ip1(net):
    return next(l for l in net.layers if l.name == "ip1")
from poseidon_amd.proto import Message, parse_text
from poseidon_amd.solver.solver import SGDSolver, get_solver


def _separable_net_param(batch=32, dim=20, classes=5, seed=3):
    """MemoryData + 2-layer MLP; data = class prototypes + small noise."""
    return parse_text("NetParameter", f"""
        name: "toy"
        layers {{ name: "data" type: MEMORY_DATA top: "data" top: "label"
                 memory_data_param {{ batch_size: {batch} channels: {dim}
                                      height: 1 width: 1 }} }}
        layers {{ name: "ip1" type: INNER_PRODUCT bottom: "data" top: "ip1"
                 blobs_lr: 1 blobs_lr: 2
                 inner_product_param {{ num_output: 32
                     weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "relu1" type: RELU bottom: "ip1" top: "ip1" }}
        layers {{ name: "ip2" type: INNER_PRODUCT bottom: "ip1" top: "ip2"
                 blobs_lr: 1 blobs_lr: 2
                 inner_product_param {{ num_output: {classes}
                     weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "loss" type: SOFTMAX_LOSS bottom: "ip2" bottom: "label"
                 top: "loss" }}
        layers {{ name: "acc" type: ACCURACY bottom: "ip2" bottom: "label"
                 top: "acc" }}
    """)


def _toy_dataset(n=256, dim=20, classes=5, seed=3):
    g = torch.Generator().manual_seed(seed)
    protos = torch.randn(classes, dim, generator=g) * 2.0
    labels = torch.randint(0, classes, (n,), generator=g)
    data = protos[labels] + 0.3 * torch.randn(n, dim, generator=g)
    return data.view(n, dim, 1, 1), labels.float()


def _solver_param(**kw):
    defaults = dict(base_lr=0.1, lr_policy="fixed", momentum=0.9,
                    weight_decay=0.0005, max_iter=60, display=0, snapshot=0)
    defaults.update(kw)
    sp = Message("SolverParameter")
    for k, v in defaults.items():
        setattr(sp, k, v)
    return sp


def test_sgd_converges_on_separable_data():
    pa.init(device="cpu", seed=7)
    sp = _solver_param()
    sp.net_param = _separable_net_param()
    solver = SGDSolver(sp, verbose=False)
    data, labels = _toy_dataset()
    solver.net.layers[0].add_data(data, labels)
    first_loss = float(solver.net.forward())
    solver.step(150)
    solver.net.forward()
    acc = float(solver.net.blobs["acc"].data)
    loss = float(solver.net.blobs["loss"].data)
    assert loss < first_loss * 0.2, (first_loss, loss)
    assert acc > 0.95, acc


@pytest.mark.parametrize("stype", ["NESTEROV", "ADAGRAD"])
def test_other_solvers_converge(stype):
    pa.init(device="cpu", seed=7)
    sp = _solver_param(base_lr=0.5 if stype == "ADAGRAD" else 0.05)
    sp.solver_type = stype
    sp.net_param = _separable_net_param()
    solver = get_solver(sp, verbose=False)
    data, labels = _toy_dataset()
    solver.net.layers[0].add_data(data, labels)
    first_loss = float(solver.net.forward())
    solver.step(150)
    solver.net.forward()
    assert float(solver.net.blobs["loss"].data) < first_loss * 0.5


def test_lr_policies():
    pa.init(device="cpu", seed=1)
    sp = _solver_param(base_lr=0.1, lr_policy="inv", gamma=0.0001, power=0.75)
    sp.net_param = _separable_net_param()
    s = SGDSolver(sp, verbose=False)
    assert s.get_learning_rate() == pytest.approx(0.1)
    s.iter = 1000
    assert s.get_learning_rate() == pytest.approx(0.1 * (1.1 ** -0.75))
    for policy, extra, iter_, want in [
        ("fixed", {}, 500, 0.1),
        ("step", {"gamma": 0.5, "stepsize": 100}, 250, 0.1 * 0.25),
        ("exp", {"gamma": 0.99}, 10, 0.1 * 0.99 ** 10),
        ("poly", {"power": 2.0, "max_iter": 1000}, 500, 0.1 * 0.25),
    ]:
        sp2 = _solver_param(base_lr=0.1, lr_policy=policy, **extra)
        sp2.net_param = _separable_net_param()
        s2 = SGDSolver(sp2, verbose=False)
        s2.iter = iter_
        assert s2.get_learning_rate() == pytest.approx(want), policy


def test_snapshot_restore(tmp_path):
    pa.init(device="cpu", seed=11)
    sp = _solver_param(snapshot_prefix=str(tmp_path / "toy"))
    sp.net_param = _separable_net_param()
    solver = SGDSolver(sp, verbose=False)
    data, labels = _toy_dataset(n=32)  # one batch -> cursor-invariant restore
    solver.net.layers[0].add_data(data, labels)
    solver.step(20)
    path = solver.snapshot()
    assert os.path.exists(path)
    w_before = _ip1(solver.net).blobs[0].data.clone()
    h_before = {i: h.clone() for i, h in solver.history.items()}

    solver2 = SGDSolver(sp, verbose=False)
    solver2.net.layers[0].add_data(data, labels)
    solver2.restore(str(tmp_path / "toy") + "_iter_20.solverstate.0.0")
    assert solver2.iter == 20
    assert torch.allclose(_ip1(solver2.net).blobs[0].data, w_before)
    for i, h in solver2.history.items():
        assert torch.allclose(h, h_before[i], atol=1e-6)
    # deterministic continuation: same data order -> same next step
    solver.step(5)
    solver2.step(5)
    assert torch.allclose(_ip1(solver.net).blobs[0].data,
                          _ip1(solver2.net).blobs[0].data, atol=1e-5)


def test_finetune_load_weights(tmp_path):
    pa.init(device="cpu", seed=13)
    sp = _solver_param(snapshot_prefix=str(tmp_path / "ft"))
    sp.net_param = _separable_net_param()
    solver = SGDSolver(sp, verbose=False)
    data, labels = _toy_dataset()
    solver.net.layers[0].add_data(data, labels)
    solver.step(10)
    model = solver.snapshot()

    solver2 = SGDSolver(sp, verbose=False)
    solver2.load_weights(model)
    assert torch.allclose(_ip1(solver2.net).blobs[0].data,
                          _ip1(solver.net).blobs[0].data)


def test_test_net_evaluation():
    pa.init(device="cpu", seed=17)
    sp = _solver_param(max_iter=50)
    sp.test_iter.append(4)
    sp.test_interval = 1000  # only explicit test() calls
    sp.test_initialization = False
    sp.net_param = _separable_net_param()
    solver = SGDSolver(sp, verbose=False)
    data, labels = _toy_dataset()
    solver.net.layers[0].add_data(data, labels)
    solver.test_nets[0].layers[0].add_data(data, labels)
    solver.step(150)
    res = solver.test(0)
    assert res["acc"] > 0.9, res
    assert res["loss"] < 0.3, res


def test_restore_preserves_test_net_sharing(tmp_path):
    """restore()/load_weights() must copy weights IN PLACE
    so test nets sharing storage via _share_params keep seeing the train
    net's tensors (reference Blob::FromProto memcpy, blob.cpp:399-426)."""
    pa.init(device="cpu", seed=23)
    sp = _solver_param(snapshot_prefix=str(tmp_path / "share"))
    sp.test_iter.append(2)
    sp.test_interval = 10**9
    sp.test_initialization = False
    sp.net_param = _separable_net_param()
    solver = SGDSolver(sp, verbose=False)
    data, labels = _toy_dataset(n=32)
    solver.net.layers[0].add_data(data, labels)
    solver.test_nets[0].layers[0].add_data(data, labels)
    solver.step(10)
    solver.snapshot()

    solver2 = SGDSolver(sp, verbose=False)
    solver2.net.layers[0].add_data(data, labels)
    solver2.test_nets[0].layers[0].add_data(data, labels)
    solver2.restore(str(tmp_path / "share") + "_iter_10.solverstate")
    assert solver2.iter == 10  # suffix-less path resolved (ADVICE low)
    # same storage objects after restore
    tr = _ip1(solver2.net).blobs[0]
    te = next(l for l in solver2.test_nets[0].layers
              if l.name == "ip1").blobs[0]
    assert tr.data.data_ptr() == te.data.data_ptr()
    # and stays shared through further training
    solver2.step(5)
    assert torch.equal(tr.data, te.data)


def test_restore_rejects_mismatched_history(tmp_path):
    pa.init(device="cpu", seed=29)
    sp = _solver_param(snapshot_prefix=str(tmp_path / "mm"))
    sp.net_param = _separable_net_param()
    solver = SGDSolver(sp, verbose=False)
    data, labels = _toy_dataset(n=32)
    solver.net.layers[0].add_data(data, labels)
    solver.step(3)
    solver.snapshot()

    sp2 = _solver_param(snapshot_prefix=str(tmp_path / "mm"))
    sp2.net_param = _separable_net_param(classes=7)  # changed prototxt
    solver2 = SGDSolver(sp2, verbose=False)
    with pytest.raises(ValueError):
        solver2.restore(str(tmp_path / "mm") + "_iter_3.solverstate.0.0")


def test_netoutputs_flattens_vector_outputs(tmp_path):
    """.netoutputs must carry every element of every net
    output (solver.cpp:336-366), not only scalar outputs."""
    pa.init(device="cpu", seed=31)
    np_param = parse_text("NetParameter", """
        name: "vec"
        layers { name: "data" type: DUMMY_DATA top: "data"
                 dummy_data_param { num: 4 channels: 3 height: 1 width: 1
                     data_filler { type: "gaussian" std: 1.0 } } }
        layers { name: "ip" type: INNER_PRODUCT bottom: "data" top: "ip"
                 inner_product_param { num_output: 3
                     weight_filler { type: "xavier" } } }
        layers { name: "sm" type: SOFTMAX bottom: "ip" top: "sm" }
    """)
    sp = _solver_param(max_iter=2, display=1)
    sp.net_param = np_param
    solver = SGDSolver(sp, verbose=False)
    solver.net.forward()
    solver._display(0.5, 0.1)
    # 'sm' output blob has 4*3 = 12 elements -> sm_0..sm_11 columns
    cols = solver._net_outputs_cols
    assert "sm_0" in cols and "sm_11" in cols, cols
    out = tmp_path / "vec"
    solver.write_net_outputs(str(out))
    header = (tmp_path / "vec.netoutputs").read_text().splitlines()[0]
    assert header.startswith("iter,time,loss,") and "sm_5" in header


def test_lenet_cpu_single_process_config1():
    """BASELINE.json config 1: LeNet on MNIST-shaped data, CPU solver,
    single process (plumbing slice). Synthetic two-class 28x28 digits
    stand in for MNIST (no datasets in the image); the zoo LeNet must
    train end-to-end and beat chance decisively."""
    import numpy as np
    from poseidon_amd.models import zoo
    pa.init(device="cpu", seed=19)
    npb = zoo.build_net("lenet", batch=32)
    # swap the data layer for MEMORY_DATA so we can feed synthetic digits
    from poseidon_amd.proto import Message
    np2 = Message("NetParameter", name="lenet_cpu")
    for lp in npb.layers:
        if lp.enum_name("type") in ("DUMMY_DATA", "DATA"):
            d = np2.add("layers", name="data", type="MEMORY_DATA")
            d.top.append("data")
            d.top.append("label")
            mp = d.ensure("memory_data_param")
            mp.batch_size = 32
            mp.channels, mp.height, mp.width = 1, 28, 28
        else:
            np2.layers.append(lp)
    sp = _solver_param(base_lr=0.01, max_iter=10**9)
    sp.net_param = np2
    solver = SGDSolver(sp, verbose=False)
    g = torch.Generator().manual_seed(4)
    n = 256
    labels = torch.randint(0, 10, (n,), generator=g)
    data = torch.zeros(n, 1, 28, 28)
    for i in range(n):  # class k = bright kth row band + noise
        r = int(labels[i]) * 2 + 4
        data[i, 0, r:r + 2, 4:24] = 1.0
    data += 0.1 * torch.randn(n, 1, 28, 28, generator=g)
    solver.net.layers[0].add_data(data, labels.float())
    first = float(solver.net.forward())
    solver.step(120)
    solver.net.forward()
    acc = float(solver.net.blobs["accuracy"].data) \
        if "accuracy" in solver.net.blobs else None
    loss = float(solver.net.blobs["loss"].data)
    assert loss < first * 0.3, (first, loss)
    if acc is not None:
        assert acc > 0.8, acc


def test_graph_capture_refused_for_non_sgd():
    """enable_graph is SGD-only by design (Nesterov/AdaGrad keep their
    per-param fused kernels eager)."""
    from poseidon_amd.solver.solver import NesterovSolver
    pa.init(device="cpu", seed=3)
    sp = _solver_param()
    sp.solver_type = "NESTEROV"
    sp.net_param = _separable_net_param()
    s = NesterovSolver(sp, verbose=False)
    assert s.enable_graph() is False
