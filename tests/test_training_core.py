import math

import torch

from automodel_amd.datasets.loader import build_dataloader, padded_collate
from automodel_amd.datasets.mock import MockDataset, MockIterableDataset
from automodel_amd.optim.adamw import FusedAdamW
from automodel_amd.optim.lr_scheduler import WarmupDecayLR
from automodel_amd.training.rng import ScopedRNG, StatefulRNG
from automodel_amd.training.step_scheduler import StepScheduler


def test_step_scheduler_grad_accum_groups():
    data = list(range(10))
    s = StepScheduler(grad_acc_steps=3, dataloader=data, max_steps=None)
    groups = list(s)
    assert [len(g) for g in groups] == [3, 3, 3]  # trailing partial dropped
    assert s.step == 3


def test_step_scheduler_max_steps_and_state():
    s = StepScheduler(grad_acc_steps=1, dataloader=list(range(100)), max_steps=5)
    assert len(list(s)) == 5 and s.finished
    state = s.state_dict()
    s2 = StepScheduler(grad_acc_steps=1, dataloader=[], max_steps=5)
    s2.load_state_dict(state)
    assert s2.step == 5


def test_step_scheduler_cadence():
    s = StepScheduler(grad_acc_steps=1, ckpt_every_steps=2, val_every_steps=3,
                      dataloader=list(range(6)))
    hits = [(s.step, s.is_ckpt_step, s.is_val_step) for _ in s]
    assert hits[1][1] and not hits[0][1]  # step 2 is ckpt step
    assert hits[2][2]                     # step 3 is val step


def test_stateful_rng_roundtrip():
    rng = StatefulRNG(seed=7)
    a = torch.randn(3)
    state = rng.state_dict()
    b = torch.randn(3)
    rng.load_state_dict(state)
    b2 = torch.randn(3)
    assert torch.equal(b, b2) and not torch.equal(a, b)


def test_scoped_rng_restores():
    torch.manual_seed(1)
    _ = torch.randn(2)
    before = torch.get_rng_state()
    with ScopedRNG(99):
        _ = torch.randn(5)
    assert torch.equal(torch.get_rng_state(), before)


def test_warmup_cosine_lr():
    m = torch.nn.Linear(2, 2)
    opt = FusedAdamW(m.parameters(), lr=1.0)
    sched = WarmupDecayLR(opt, warmup_steps=10, total_steps=110, decay="cosine",
                          min_lr_ratio=0.1)
    assert opt.param_groups[0]["lr"] < 1.0  # warmup start
    for _ in range(10):
        sched.step()
    assert abs(opt.param_groups[0]["lr"] - 1.0) < 0.11
    for _ in range(100):
        sched.step()
    assert abs(opt.param_groups[0]["lr"] - 0.1) < 0.02  # decayed to min ratio


def test_adamw_cpu_matches_torch():
    torch.manual_seed(0)
    p1 = torch.nn.Parameter(torch.randn(32))
    p2 = torch.nn.Parameter(p1.detach().clone())
    mine = FusedAdamW([p1], lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.1)
    ref = torch.optim.AdamW([p2], lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.1)
    for i in range(5):
        g = torch.randn(32)
        p1.grad = g.clone()
        p2.grad = g.clone()
        mine.step()
        ref.step()
        assert torch.allclose(p1, p2, atol=1e-6), (i, (p1 - p2).abs().max())


def test_padded_collate():
    batch = [
        {"input_ids": torch.tensor([1, 2, 3]), "labels": torch.tensor([2, 3, 4])},
        {"input_ids": torch.tensor([1]), "labels": torch.tensor([2])},
    ]
    out = padded_collate(batch, pad_token_id=0)
    assert out["input_ids"].shape == (2, 3)
    assert out["labels"][1, 1] == -100


def test_mock_datasets_and_loader():
    ds = MockDataset(num_samples=8, seq_len=16, vocab_size=100)
    loader = build_dataloader(ds, batch_size=2, shuffle=False)
    batches = list(loader)
    assert len(batches) == 4
    assert batches[0]["input_ids"].shape == (2, 16)

    it = MockIterableDataset(seq_len=8, vocab_size=50, num_samples=6)
    loader = build_dataloader(it, batch_size=2)
    assert len(list(loader)) == 3


def test_dataloader_dp_sharding_disjoint():
    ds = MockDataset(num_samples=8, seq_len=4)
    l0 = build_dataloader(ds, batch_size=1, shuffle=False, dp_rank=0, dp_world=2)
    l1 = build_dataloader(ds, batch_size=1, shuffle=False, dp_rank=1, dp_world=2)
    ids0 = torch.cat([b["input_ids"] for b in l0])
    ids1 = torch.cat([b["input_ids"] for b in l1])
    assert len(ids0) == 4 and len(ids1) == 4
    assert not torch.equal(ids0, ids1)


def test_timers():
    import time as _t

    from automodel_amd.training.timers import Timers

    tm = Timers(cuda_sync=False)
    with tm("a"):
        _t.sleep(0.01)
    assert tm.mean("a") >= 0.009
    assert "a" in tm.summary()


def test_capabilities_validation():
    from automodel_amd.models.common.capabilities import validate_model_against_mesh
    from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM

    m = LlamaForCausalLM(LlamaConfig(vocab_size=64, hidden_size=32,
                                     intermediate_size=64, num_hidden_layers=1,
                                     num_attention_heads=3, num_key_value_heads=3,
                                     max_position_embeddings=32))
    assert validate_model_against_mesh(m, {"tp": 1}) == []
    assert validate_model_against_mesh(m, {"tp": 2})  # 3 heads not divisible


def test_slurm_launcher_render(tmp_path):
    from automodel_amd.launcher.slurm import SlurmLauncher

    l = SlurmLauncher(nodes=2, gpus_per_node=8, account="acct")
    script = l.render("cfg.yaml", "pkg.Recipe", ["--a.b=1"])
    assert "--nnodes=2" in script and "--nproc-per-node=8" in script
    assert "HSA_ENABLE_IPC_MODE_LEGACY=0" in script
    assert "#SBATCH --account=acct" in script
    path = l.launch("cfg.yaml", "pkg.Recipe", [], script_path=str(tmp_path / "j.sub"),
                    submit=False)
    assert (tmp_path / "j.sub").exists()


def test_async_checkpoint_writer(tmp_path):
    import torch as _torch

    from automodel_amd.checkpoint.async_save import AsyncCheckpointWriter

    w = AsyncCheckpointWriter()
    state = {"w": _torch.randn(4, 4), "nested": {"b": _torch.ones(2)}}
    written = {}

    def write_fn(staged):
        written.update(staged)

    w.save_async(state, write_fn)
    w.wait()
    assert _torch.equal(written["w"], state["w"])
    assert written["w"].device.type == "cpu"


def test_qwen3_qk_norm_model():
    from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM

    cfg = LlamaConfig.from_hf_config({
        "architectures": ["Qwen3ForCausalLM"], "vocab_size": 64,
        "hidden_size": 32, "intermediate_size": 64, "num_hidden_layers": 1,
        "num_attention_heads": 2, "num_key_value_heads": 1,
        "max_position_embeddings": 32,
    })
    assert cfg.qk_norm
    m = LlamaForCausalLM(cfg)
    m.init_weights()
    assert hasattr(m.model.layers[0].self_attn, "q_norm")
    out = m(torch.randint(0, 64, (1, 8)))
    assert out.shape == (1, 8, 64)


def test_skypilot_launcher_renders_task(tmp_path):
    from automodel_amd.launcher.skypilot import SkyPilotConfig, SkyPilotLauncher

    l = SkyPilotLauncher(cloud="kubernetes", accelerators="MI355X:8", num_nodes=2,
                         job_name="ft")
    task = l.render_task("cfg.yaml", "automodel_amd.recipes.llm.train_ft",
                         ["--optimizer.lr=1e-4"])
    assert task["num_nodes"] == 2 and task["resources"]["accelerators"] == "MI355X:8"
    assert task["envs"]["HSA_ENABLE_IPC_MODE_LEGACY"] == "0"
    assert "--nproc-per-node=8" in task["run"]
    assert "--node-rank=$SKYPILOT_NODE_RANK" in task["run"]
    assert "--optimizer.lr=1e-4" in task["run"]

    # single node -> standalone rendezvous
    single = SkyPilotLauncher(accelerators="MI355X:4").render_task("c.yaml", "t")
    assert "--standalone" in single["run"]

    # task file written, valid YAML, no submission without `sky`
    p = l.launch("cfg.yaml", "t", task_path=str(tmp_path / "task.yaml"), submit=False)
    import yaml as _y

    loaded = _y.safe_load(open(p))
    assert loaded["name"] == "ft"

    import pytest as _pt

    with _pt.raises(ValueError):
        SkyPilotConfig(cloud="notacloud")
