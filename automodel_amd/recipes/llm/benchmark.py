"""Benchmarking recipe: mock data, warmup, per-step timing, rocprof ranges.

Reference behavior: nemo_automodel/recipes/llm/benchmark.py:102-615
(BenchmarkingRecipeForNextTokenPrediction: swaps in MockIterableDataset,
does warmup steps, times each optimizer step, drives the profiler between
configured steps via cudaProfilerStart/Stop — here rocTX ranges through
torch.cuda.nvtx, which maps onto roctracer on ROCm).
"""

from __future__ import annotations

import json
import sys
import time

import torch

from automodel_amd.config.loader import ConfigNode, apply_overrides, load_yaml_config, parse_cli_overrides
from automodel_amd.recipes.llm.train_ft import TrainFinetuneRecipeForNextTokenPrediction
from automodel_amd.utils.flops import MI355X_PEAK_BF16, llama_flops_per_token


class BenchmarkingRecipeForNextTokenPrediction(TrainFinetuneRecipeForNextTokenPrediction):
    def setup(self) -> None:
        bench_cfg = self.cfg.get("benchmark", ConfigNode())
        # force mock iterable data of the configured shape
        seq_len = bench_cfg.get("seq_len", 4096)
        self.cfg.set_by_dotted("dataloader.dataset.kind", "mock_iterable")
        self.cfg.set_by_dotted("dataloader.dataset.seq_len", seq_len)
        self.cfg.set_by_dotted(
            "dataloader.dataset.vocab_size",
            self.cfg.get_by_dotted("model.config.vocab_size", 128256),
        )
        super().setup()
        self.warmup_steps = bench_cfg.get("warmup_steps", 3)
        self.profile_start = bench_cfg.get("profile_start_step", -1)
        self.profile_stop = bench_cfg.get("profile_stop_step", -1)
        self.step_times: list[float] = []
        self.seq_len = seq_len

    def run_train_validation_loop(self) -> None:
        self.model.train()
        use_cuda = torch.cuda.is_available()
        for batches in self.step_scheduler:
            step = self.step_scheduler.step
            if step == self.profile_start and use_cuda:
                torch.cuda.nvtx.range_push("benchmark_profile")
            if use_cuda:
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            metrics = self._run_train_optim_step(batches)
            if use_cuda:
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            if step == self.profile_stop and use_cuda:
                torch.cuda.nvtx.range_pop()
            if step > self.warmup_steps:
                self.step_times.append(dt)
            self.metrics.log({**metrics, "bench_step_time_s": dt})
            if self.rank == 0:
                self.logger.info(f"bench step {step}: {dt*1e3:.1f} ms")
        self._report()

    def _report(self) -> None:
        if self.rank != 0 or not self.step_times:
            return
        import statistics

        mean_t = statistics.mean(self.step_times)
        cfg = self.model.config
        tokens_per_step = (
            self.cfg.get_by_dotted("dataloader.batch_size", 1)
            * self.step_scheduler.grad_acc_steps
            * self.seq_len
            * self.mesh.dp_size
        )
        if hasattr(cfg, "moe"):
            from automodel_amd.utils.flops import moe_flops_per_token
            fpt = moe_flops_per_token(
                cfg.hidden_size, cfg.num_hidden_layers, cfg.vocab_size,
                self.seq_len, cfg.num_attention_heads, cfg.num_key_value_heads,
                cfg.moe.moe_intermediate_size or cfg.intermediate_size,
                cfg.moe.n_activated_experts, cfg.moe.n_shared_experts,
                cfg.moe.shared_expert_intermediate_size, cfg.head_dim,
            )
        else:
            fpt = llama_flops_per_token(
                cfg.hidden_size, cfg.intermediate_size, cfg.num_hidden_layers,
                cfg.vocab_size, self.seq_len, cfg.num_attention_heads,
                cfg.num_key_value_heads, cfg.head_dim,
            )
        tps = tokens_per_step / mean_t
        summary = {
            "mean_step_time_s": round(mean_t, 4),
            "tokens_per_sec": round(tps, 1),
            "tokens_per_sec_per_gpu": round(tps / max(1, self.world), 1),
            "tflops_per_sec_per_gpu": round(tps / max(1, self.world) * fpt / 1e12, 1),
            "mfu": round(tps / max(1, self.world) * fpt / MI355X_PEAK_BF16, 4),
            "n_steps_timed": len(self.step_times),
        }
        self.logger.info("benchmark summary: " + json.dumps(summary))
        self.metrics.log({"benchmark_summary": summary})


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    r = BenchmarkingRecipeForNextTokenPrediction(cfg)
    r.setup()
    r.run_train_validation_loop()


if __name__ == "__main__":
    main()
