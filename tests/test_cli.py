"""CLI-level integration: train.py (fresh + --resume) and the CBF contour
plotting path, all on CPU with tiny configs."""
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_train_cli_and_resume(tmp_path):
    env = dict(os.environ)
    common = [sys.executable, "train.py", "--env", "DoubleIntegrator", "-n", "3",
              "--area-size", "2", "--cpu", "--steps", "2", "--n-env-train", "2",
              "--n-env-test", "2", "--horizon", "2", "--buffer-size", "8",
              "--eval-interval", "1", "--save-interval", "1",
              "--log-dir", str(tmp_path), "--seed", "7"]
    r = subprocess.run(common, cwd=ROOT, env=env, capture_output=True, text=True,
                       timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    runs = os.listdir(tmp_path / "DoubleIntegrator" / "gcbf+")
    assert len(runs) == 1
    run = tmp_path / "DoubleIntegrator" / "gcbf+" / runs[0]
    assert (run / "models" / "2" / "cbf.pkl").exists()
    assert (run / "models" / "resume.pt").exists()
    assert (run / "config.yaml").exists()

    # resume from the saved full state for 1 more step
    r2 = subprocess.run(common + ["--steps", "3", "--resume",
                                  str(run / "models" / "resume.pt")],
                        cwd=ROOT, env=env, capture_output=True, text=True,
                        timeout=420)
    assert r2.returncode == 0, r2.stderr[-2000:]
    assert "Resumed full training state" in r2.stdout


def test_plot_cbf_contours(tmp_path):
    """trainer.utils.plot_cbf renders the h contour overlay (reference
    trainer/utils.py:112-146 / test.py --cbf path)."""
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.algo import make_algo
    from gcbfplus_amd.trainer.utils import plot_cbf

    env = make_env("DoubleIntegrator", 3, area_size=2.0, max_step=4, device="cpu")
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim, n_agents=3)
    g = env.reset(1, np.random.default_rng(0))
    fig = plt.figure()
    cbf = lambda gr: algo.get_cbf(gr)
    out = plot_cbf(fig, cbf, env, g, agent_id=0, n_mesh=8)
    p = tmp_path / "cbf.png"
    out.savefig(p)
    assert p.exists() and p.stat().st_size > 1000
    plt.close(out)
