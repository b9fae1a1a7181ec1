"""BASELINE config #2 through the CLI: the pytorch-1gpu recipe's bf16
matmul task pinned to 1 MI355X via the ROCm binder."""
from pathlib import Path

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

from click.testing import CliRunner  # noqa: E402

from shipyard_amd.cli import cli  # noqa: E402

RECIPES = Path(__file__).parents[1] / "recipes"


def test_pytorch_1gpu_recipe(tmp_path):
    assert torch.cuda.is_available()
    runner = CliRunner()

    def run(args):
        return runner.invoke(
            cli, args + ["--configdir", str(RECIPES / "pytorch-1gpu"),
                         "--root", str(tmp_path / "root")],
            catch_exceptions=False)

    r = run(["pool", "add"])
    assert r.exit_code == 0, r.output
    r = run(["jobs", "add", "--wait"])
    assert r.exit_code == 0, r.output
    r = run(["data", "files", "stream",
             "--filespec", "torch-matmul,bf16-matmul"])
    assert "bf16 matmul ok" in r.output, r.output
    r = run(["jobs", "stats", "--jobid", "torch-matmul"])
    assert '"completed": 1' in r.output


def test_tiny_gpt2_recipe(tmp_path):
    """The ML-framework recipe family analogue: random-init tiny GPT-2
    training steps through the executor on one MI355X."""
    assert torch.cuda.is_available()
    runner = CliRunner()

    def run(args):
        return runner.invoke(
            cli, args + ["--configdir",
                         str(RECIPES / "transformers-tiny-gpt2"),
                         "--root", str(tmp_path / "root")],
            catch_exceptions=False)

    r = run(["pool", "add"])
    assert r.exit_code == 0, r.output
    r = run(["jobs", "add", "--wait"])
    assert r.exit_code == 0, r.output
    r = run(["data", "files", "stream", "--filespec", "tiny-gpt2,train-step"])
    assert "tokens/s" in r.output, r.output


def test_gemm_burn_recipe(tmp_path):
    """HPLinpack analogue: sustained bf16 GEMM burn, >100 TFLOP/s
    sanity floor (MI355X dense bf16 peak ~2.5 PFLOP/s)."""
    assert torch.cuda.is_available()
    runner = CliRunner()

    def run(args):
        return runner.invoke(
            cli, args + ["--configdir", str(RECIPES / "gemm-burn-hpl"),
                         "--root", str(tmp_path / "root")],
            catch_exceptions=False)

    r = run(["pool", "add"])
    assert r.exit_code == 0, r.output
    r = run(["jobs", "add", "--wait"])
    assert r.exit_code == 0, r.output
    r = run(["data", "files", "stream", "--filespec", "gemm-burn,burn"])
    assert "TFLOP/s" in r.output, r.output
    tflops = float(r.output.split("bf16 GEMM")[1].split("TFLOP/s")[0])
    assert tflops > 100, r.output
