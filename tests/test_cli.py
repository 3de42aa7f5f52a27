"""End-to-end CLI tests: train.py -> checkpoint -> test.py round trip on CPU
(BASELINE config 1: SimpleCar n=4, GCBF, short run)."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(cmd, timeout=900, **kw):
    return subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                          timeout=timeout, **kw)


@pytest.fixture(scope="module")
def trained_run(tmp_path_factory):
    log_root = str(tmp_path_factory.mktemp("logs"))
    r = _run([sys.executable, "train.py", "--env", "SimpleCar", "-n", "4",
              "--steps", "60", "--batch-size", "20", "--cpu",
              "--log-path", log_root, "--eval-epi", "0", "--seed", "0"])
    assert r.returncode == 0, r.stderr[-3000:]
    run_dirs = os.listdir(os.path.join(log_root, "SimpleCar", "gcbf"))
    assert len(run_dirs) == 1
    return os.path.join(log_root, "SimpleCar", "gcbf", run_dirs[0])


def test_train_writes_layout(trained_run):
    # reference checkpoint layout: models/step_N/{cbf.pkl, actor.pkl}
    assert os.path.exists(os.path.join(trained_run, "settings.yaml"))
    models = os.listdir(os.path.join(trained_run, "models"))
    assert any(m.startswith("step_") for m in models)
    step_dir = os.path.join(trained_run, "models", sorted(models)[-1])
    assert os.path.exists(os.path.join(step_dir, "cbf.pkl"))
    assert os.path.exists(os.path.join(step_dir, "actor.pkl"))
    # scalar logs
    assert os.path.exists(os.path.join(trained_run, "summary",
                                       "scalars.jsonl"))


def test_settings_roundtrip(trained_run):
    from gcbf_amd.trainer.utils import read_settings
    s = read_settings(trained_run)
    assert s["env"] == "SimpleCar"
    assert s["num_agents"] == 4
    assert s["algo"] == "gcbf"
    assert s["hyper_params"]["inner_iter"] == 10


def test_test_cli_loads_checkpoint(trained_run):
    # --rand 0: with a barely-trained policy the refinement noise can make
    # agents wander for the whole 2500-step episode cap on CPU
    r = _run([sys.executable, "test.py", "--path", trained_run, "--epi", "1",
              "--no-video", "--cpu", "--rand", "0"], timeout=1800)
    assert r.returncode == 0, r.stderr[-3000:]
    assert "safe rate" in r.stdout
    assert os.path.exists(os.path.join(trained_run, "test_log.csv"))


def test_resume_flag(trained_run):
    r = _run([sys.executable, "train.py", "--env", "SimpleCar", "-n", "4",
              "--steps", "80", "--batch-size", "20", "--cpu",
              "--eval-epi", "0", "--resume", trained_run])
    assert r.returncode == 0, r.stderr[-3000:]
    assert "Resuming" in r.stdout


def test_bench_contract():
    r = _run([sys.executable, "bench.py", "--gpus", "1", "--steps", "24",
              "--warmup", "20", "--batch-size", "20", "-n", "4"],
             timeout=900)
    assert r.returncode == 0, r.stderr[-3000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    j = json.loads(line)
    assert j["metric"] == "env-steps/sec (whole node)"
    assert j["n_gpus"] == 1 and j["steps"] == 24 and j["warmup"] == 20
    assert j["value"] > 0 and j["higher_is_better"] is True
    assert j["scaling"] == "weak" and j["data"] == "synthetic"
    assert "ms_per_step" in j and "config" in j


def test_bench_refuses_mislabeled_world():
    """Driver contract guard: WORLD_SIZE inconsistent with --gpus must
    exit non-zero instead of printing mislabeled numbers."""
    import os
    env = dict(os.environ)
    env.update({"WORLD_SIZE": "2", "RANK": "0", "LOCAL_RANK": "0",
                "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29400"})
    r = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "4",
         "--warmup", "1", "--batch-size", "8", "-n", "4"],
        capture_output=True, text=True, timeout=300, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 2, (r.returncode, r.stdout, r.stderr[-500:])
    assert "refusing" in r.stdout


def test_nominal_test_cli(tmp_path):
    r = _run([sys.executable, "test.py", "--env", "SimpleCar", "-n", "3",
              "--epi", "1", "--no-video", "--cpu"], timeout=1800)
    assert r.returncode == 0, r.stderr[-3000:]


def test_plot_cbf_cli(trained_run):
    r = _run([sys.executable, "plot_cbf.py", "--path", trained_run,
              "--area-size", "4.0", "--epi", "1", "--max-steps", "2",
              "--cpu"], timeout=1800)
    assert r.returncode == 0, r.stderr[-3000:]
    agent_dir = os.path.join(trained_run, "figs", "agent_0", "epi_0")
    assert os.path.exists(os.path.join(agent_dir, "0.pdf"))
