"""Outer training loop (reference trainer/trainer.py:18-143)."""
from __future__ import annotations

import os
import time
from typing import Optional

import numpy as np
import torch

from .trainer_utils_compat import tqdm
from .utils import MetricsLogger, collect_rollout, eval_rollout_metrics


class Trainer:
    def __init__(
        self,
        env,
        env_test,
        algo,
        n_env_train: int,
        n_env_test: int,
        log_dir: Optional[str],
        seed: int,
        params: dict,
        save_log: bool = True,
    ):
        self.env = env
        self.env_test = env_test
        self.algo = algo
        self.n_env_train = n_env_train
        self.n_env_test = n_env_test
        self.log_dir = log_dir
        self.seed = seed
        self.params = self._check_params(params)
        self.save_log = save_log and log_dir is not None

        if self.save_log:
            self.model_dir = os.path.join(log_dir, "models")
            os.makedirs(self.model_dir, exist_ok=True)
        self.logger = MetricsLogger(log_dir if self.save_log else None,
                                    run_name=params.get("run_name", "run"))

        self.steps = params["training_steps"]
        self.eval_interval = params["eval_interval"]
        self.eval_epi = params["eval_epi"]
        self.save_interval = params["save_interval"]
        self.update_steps = 0
        self.rng = np.random.default_rng(seed)
        self.test_rng = np.random.default_rng(seed)
        self._ro_train = None
        self._ro_test = None
        if torch.cuda.is_available() and env.device.type == "cuda":
            from .graphing import GraphedRolloutStep

            self._ro_train = GraphedRolloutStep(env, algo.step)
            self._ro_test = GraphedRolloutStep(env_test, algo.act)

    @staticmethod
    def _check_params(params: dict) -> dict:
        for k in ("run_name", "training_steps", "eval_interval", "eval_epi", "save_interval"):
            assert k in params, f"{k} not found in params"
        assert params["eval_interval"] > 0 and params["eval_epi"] >= 1
        assert params["save_interval"] > 0
        return params

    def train(self, start_step: int = 0):
        """Run the outer loop; ``start_step`` > 0 resumes a run restored via
        ``algo.load_full`` (framework extra over the reference — SURVEY §5.4:
        the reference checkpoints params only)."""
        start = time.time()
        pbar = tqdm(total=self.steps, ncols=80)
        pbar.update(start_step)
        for step in range(start_step, self.steps + 1):
            if step % self.eval_interval == 0:
                eval_info = self.eval_step()
                self.logger.log({**eval_info, "step": step}, step=self.update_steps)
                el = time.time() - start
                tqdm.write(
                    f"step: {step:4d}, time: {el:5.0f}s, reward: {eval_info['eval/reward']:9.4f}, "
                    f"cost: {eval_info['eval/cost']:8.4f}, "
                    f"unsafe_frac: {eval_info['eval/unsafe_frac']:6.2f}, "
                    f"finish: {eval_info['eval/finish']:6.2f}"
                )
                if self.save_log and step % self.save_interval == 0:
                    self.algo.save(self.model_dir, step)
                    # full training state for --resume (optimizers, target
                    # net, rng) — overwritten in place each save
                    self.algo.save_full(os.path.join(self.model_dir, "resume.pt"), step)

            graph0 = self.env.reset(self.n_env_train, self.rng)
            rollout = collect_rollout(self.env, self.algo.step, graph0, self._ro_train)
            update_info = self.algo.update(rollout, step)
            self.logger.log(update_info, step=self.update_steps)
            self.update_steps += 1
            pbar.update(1)
        pbar.close()
        self.logger.close()

    def eval_step(self) -> dict:
        # Fixed eval worlds: the reference pre-splits one set of test keys
        # before the loop (trainer/trainer.py:99-100) so every eval interval
        # measures the same scenarios — re-seed per eval instead of advancing
        # self.test_rng.
        rng = np.random.default_rng(self.seed)
        graph0 = self.env_test.reset(self.n_env_test, rng)
        rollout = collect_rollout(self.env_test, self.algo.act, graph0, self._ro_test)
        return eval_rollout_metrics(self.env_test, rollout)
