"""CLI surface tests: chained operators, flags, plugin op, task stream."""
import os
import re

import numpy as np
import pytest
import torch
from click.testing import CliRunner

from chunkflow_amd.flow import main

GPU = torch.cuda.is_available()


def run_cli(args):
    runner = CliRunner()
    result = runner.invoke(main, args, catch_exceptions=False)
    assert result.exit_code == 0, result.output
    return result


def test_generate_tasks_stream():
    r = run_cli(['generate-tasks', '--roi-size', '512', '1024', '2048',
                 '--chunk-size', '512', '512', '512'])
    assert 'total number of tasks: 8' in r.output


def test_create_chunk_save_npy(tmp_path):
    out = tmp_path / 'c.npy'
    run_cli(['create-chunk', '--size', '16', '24', '32', '--dtype', 'uint8',
             '--pattern', 'sin', 'save-npy', '-f', str(out)])
    arr = np.load(out)
    assert arr.shape == (16, 24, 32) and arr.dtype == np.uint8


@pytest.mark.skipif(GPU, reason='CPU plumbing pipeline; GPU covered in -m gpu')
def test_cpu_pipeline_config1_shape(tmp_path, golden):
    """BASELINE config 1 (plumbing, no GPU): create-chunk -> inference
    (identity) -> crop-margin -> save-npy, small geometry."""
    _, arrays = golden
    out = tmp_path / 'out.npy'
    run_cli(['create-chunk', '--size', '20', '68', '72',
             '--dtype', 'uint8', '--pattern', 'sin',
             'inference', '-s', '10', '32', '32',
             '--output-patch-overlap', '2', '8', '8',
             '--framework', 'identity', '--batch-size', '3',
             '--num-output-channels', '3', '--mask-output-chunk',
             'crop-margin', '-m', '1', '2', '3', '1', '2', '3',
             'save-npy', '-f', str(out)])
    got = np.load(out)
    assert got.shape == (3, 18, 64, 66)
    sin = arrays['sin_20x68x72_u8'].astype(np.float32) / 255.0
    np.testing.assert_allclose(got[0], sin[1:-1, 2:-2, 3:-3],
                               rtol=1e-5, atol=1e-5)


@pytest.mark.skipif(GPU, reason='CPU plumbing pipeline')
def test_cpu_pipeline_normalize_plugin_cc(tmp_path):
    """normalize-intensity + plugin + connected-components chain on CPU."""
    plugin = tmp_path / 'clip01.py'
    plugin.write_text(
        'import numpy as np\n'
        'def execute(chunk):\n'
        '    return np.clip(np.asarray(chunk.array), 0, 1)'
        '.astype(np.float32)\n')
    out = tmp_path / 'seg.npy'
    run_cli(['create-chunk', '--size', '12', '20', '24', '--dtype', 'uint8',
             'normalize-intensity',
             'plugin', '-f', str(plugin), '-i', 'chunk', '-o', 'chunk',
             'connected-components', '-t', '0.5', '-c', '6',
             'save-npy', '-f', str(out)])
    seg = np.load(out)
    assert seg.dtype == np.uint32
    assert seg.max() >= 1


def test_dry_run_inference():
    r = run_cli(['--dry-run', 'create-chunk', '--size', '20', '68', '72',
                 'inference', '-s', '10', '32', '32',
                 '--output-patch-overlap', '2', '8', '8',
                 '--framework', 'identity', '--mask-output-chunk'])
    assert r.exit_code == 0


def test_log_summary(tmp_path):
    logdir = tmp_path / 'log'
    logdir.mkdir()
    (logdir / 'a.json').write_text(
        '{"timer": {"inference": 2.0, "crop-margin": 0.5}}')
    (logdir / 'b.json').write_text(
        '{"timer": {"inference": 4.0, "crop-margin": 0.7}}')
    r = run_cli(['log-summary', '-l', str(logdir),
                 '-s', '512', '512', '512'])
    assert 'inference: mean=3.000s' in r.output
    assert re.search(r'speed: .* kv/s', r.output)


def test_skip_family_and_var_ops(tmp_path):
    """The reference's at-least-once idempotency/resume family
    (SURVEY.md §5): skip-task-by-file, mark-complete, skip-all-zero,
    copy-var, delete-var."""
    mark = str(tmp_path / 'done_')
    # first run marks completion for both tasks
    run_cli(['generate-tasks', '--roi-size', '8', '8', '16',
             '--chunk-size', '8', '8', '8',
             'create-chunk', '--dtype', 'uint8', '--pattern', 'sin',
             'copy-var', '-f', 'chunk', '-t', 'backup',
             'delete-var', '-v', 'backup',
             'mark-complete', '-p', mark])
    import glob
    assert len(glob.glob(mark + '*')) == 2
    # second run skips both tasks before create-chunk (skip-task-by-file)
    r = run_cli(['generate-tasks', '--roi-size', '8', '8', '16',
                 '--chunk-size', '8', '8', '8',
                 'skip-task-by-file', '-p', mark, '-m', 'exist',
                 'create-chunk', '--dtype', 'uint8'])
    assert 'creating chunk' in r.output  # op ran, but tasks were None

    # skip-all-zero drops zero chunks and touches a trace file
    trace = str(tmp_path / 'zero_')
    run_cli(['generate-tasks', '--roi-size', '8', '8', '8',
             '--chunk-size', '8', '8', '8',
             'create-chunk', '--dtype', 'uint8', '--pattern', 'zero',
             'skip-all-zero', '-p', trace,
             'mark-complete', '-p', str(tmp_path / 'never_')])
    assert len(glob.glob(trace + '*')) == 1
    assert len(glob.glob(str(tmp_path / 'never_*'))) == 0


@pytest.mark.skipif(GPU, reason='CPU plumbing pipeline')
def test_universal_example_file(golden):
    """The shipped universal example file works through the CLI (the
    reference's examples/inference/universal_identity.py contract)."""
    import os
    _, arrays = golden
    example = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        'examples', 'inference', 'universal_identity.py')
    import tempfile
    with tempfile.TemporaryDirectory() as td:
        out = os.path.join(td, 'u.npy')
        run_cli(['create-chunk', '--size', '20', '68', '72',
                 '--dtype', 'uint8', '--pattern', 'sin',
                 'inference', '-m', example, '-s', '10', '32', '32',
                 '--output-patch-overlap', '2', '8', '8',
                 '--framework', 'universal', '--batch-size', '3',
                 '--num-output-channels', '3', '--mask-output-chunk',
                 'save-npy', '-f', out])
        got = np.load(out)
    from chunkflow_amd.chunk import Chunk as C
    sin = C.create(size=(20, 68, 72), dtype='uint8', pattern='sin').array
    np.testing.assert_allclose(got[0], sin.astype(np.float32) / 255.0,
                               rtol=1e-5, atol=1e-5)


def test_inference_dtype_bfloat16(tmp_path, golden_dir, golden):
    """--dtype bfloat16 (documented MI355X extension of the reference's
    float32|float16 choice): bf16 compute, f32 blended output."""
    out = tmp_path / 'out.npy'
    run_cli(['create-chunk', '--size', '20', '68', '72',
             '--dtype', 'uint8', '--pattern', 'sin',
             'inference', '-s', '10', '32', '32',
             '--output-patch-overlap', '2', '8', '8',
             '--framework', 'pytorch', '--batch-size', '1',
             '--dtype', 'bfloat16',
             '-m', os.path.join(golden_dir, 'ref_model.py'),
             '-w', os.path.join(golden_dir, 'ref_model_weights.pt'),
             '--num-output-channels', '3', '--mask-output-chunk',
             'save-npy', '-f', str(out)])
    got = np.load(out)
    assert got.shape == (3, 20, 68, 72) and got.dtype == np.float32
    # bf16 compute through the 2-layer conv agrees with the committed f32
    # reference output to bf16 accumulation error (loose per-element bound,
    # tight in the mean)
    ref = golden[1]['e2e_pytorch_out']
    np.testing.assert_allclose(got, ref, rtol=0.15, atol=0.06)
    assert float(np.abs(got - ref).mean()) < 2e-2


def test_stitch_requires_bbox_tasks(tmp_path):
    """stitch without generate-tasks (no bbox on the task) fails loudly."""
    from click.testing import CliRunner
    from chunkflow_amd.flow import main
    r = CliRunner().invoke(main, [
        'create-chunk', '--size', '8', '8', '8',
        'stitch', '--backend', 'gloo',
        '-f', str(tmp_path / 'x.npy')])
    assert r.exit_code != 0
    assert isinstance(r.exception, ValueError)
