"""The `chunkflow` CLI: chained operators over a task stream.

Keeps the reference's command and flag surface verbatim for the hot-path
operators (reference chunkflow/flow/flow.py):
  generate-tasks        :73-183   (local grid decomposition; cloud/SQS
                                   distribution is replaced by in-node
                                   multi-GPU dispatch — see dispatch.py)
  create-chunk          :652-678
  normalize-intensity   :1650-1669
  inference             :1852-1933
  crop-margin           :2053-2084
  plugin                :1751-1800
  connected-components  :1803-1829 (gfx950 union-find on device chunks)
  normalize-contrast    :1672-1710 (device histogram/LUT kernels)
  load-h5 / save-h5     :976-1120  (in-repo HDF5 codec — h5io.py)
  load-tif / save-tif   :918-974   (in-repo TIFF codec — tiffio.py)
  log-summary           :1633-1647 (voxels/sec report shape)
plus the at-least-once resume family (skip-task-by-file, mark-complete,
skip-all-zero, skip-none — SURVEY.md §5), copy-var/delete-var, and npy
load/save at the pipeline edge.
"""
import json
import os
from time import time

import click
import numpy as np

from .cartesian import BoundingBox, BoundingBoxes
from .chunk import Chunk
from .plugin import Plugin
from .runtime import (DEFAULT_CHUNK_NAME, default_none, generator,
                      get_initial_task, main, operator, state)


@main.command('generate-tasks')
@click.option('--roi-start', '-s', type=click.INT, default=None, nargs=3,
              callback=default_none, help='(z y x), start of the chunks')
@click.option('--roi-stop', '-r', type=click.INT, nargs=3, default=None,
              callback=default_none, help='stop coordinate of region of interest')
@click.option('--roi-size', '-z', type=click.INT, nargs=3, default=None,
              callback=default_none, help='size of region of interest')
@click.option('--chunk-size', '-c', type=click.INT, default=None, nargs=3,
              help='(z y x), size/shape of chunks')
@click.option('--bounding-box', '-b', type=str, default=None,
              help='the string representation of a bounding box')
@click.option('--grid-size', '-g', type=click.INT, default=None, nargs=3,
              callback=default_none, help='(z y x), grid size of output blocks')
@click.option('--file-path', '-f', default=None, type=str,
              help='output tasks as an numpy array formated as npy.')
@click.option('--respect-chunk-size/--respect-stop', default=True,
              help='for the last bounding box, make the chunk size '
                   'consistent or cut off at the stopping boundary.')
@click.option('--task-index-start', '-i', type=click.INT, default=0,
              help='starting index of task list.')
@click.option('--task-index-stop', '-p', type=click.INT, default=None,
              help='stop index of task list.')
@click.option('--task-rank', type=click.INT, default=None,
              help='worker rank for multi-GPU sharding (tasks with index % '
                   'task-world == task-rank are kept). Defaults to the RANK '
                   'env var under torchrun — the slurm-array-index analog '
                   'of reference flow.py:151-156.')
@click.option('--task-world', type=click.INT, default=None,
              help='number of workers for multi-GPU sharding. Defaults to '
                   'the WORLD_SIZE env var under torchrun.')
@generator
def generate_tasks(roi_start, roi_stop, roi_size, chunk_size, bounding_box,
                   grid_size, file_path, respect_chunk_size,
                   task_index_start, task_index_stop, task_rank, task_world):
    """Generate a batch of tasks."""
    if bounding_box is not None:
        bboxes = [BoundingBox.from_string(bounding_box)]
        if chunk_size is None:
            chunk_size = tuple(bboxes[0].shape)
    else:
        bboxes = BoundingBoxes.from_manual_setup(
            chunk_size, roi_start=roi_start, roi_stop=roi_stop,
            roi_size=roi_size, grid_size=grid_size,
            respect_chunk_size=respect_chunk_size)
    if task_index_start:
        if task_index_stop is None:
            task_index_stop = len(bboxes)
        bboxes = bboxes[task_index_start:task_index_stop]
    if file_path:
        arr = np.array([list(b.start) + list(b.stop) for b in bboxes],
                       dtype=np.int64)
        np.save(file_path, arr)
    # rank-aware sharding (torchrun worker mode, BASELINE config 3):
    # every rank computes the SAME full bbox list (kept in state for the
    # stitch operator), then keeps indices i % world == rank
    if task_rank is None:
        task_rank = int(os.environ.get('RANK', '0'))
    if task_world is None:
        task_world = int(os.environ.get('WORLD_SIZE', '1'))
    assert 0 <= task_rank < task_world
    state['all_bboxes'] = list(bboxes)
    bbox_num = len(bboxes)
    print(f'total number of tasks: {bbox_num}')
    for bbox_index, bbox in enumerate(bboxes):
        if bbox_index % task_world != task_rank:
            continue
        task = get_initial_task()
        task['bbox'] = bbox
        task['bbox_index'] = bbox_index
        task['bbox_num'] = bbox_num
        task['log']['bbox'] = bbox.string
        yield task


@main.command('create-chunk')
@click.option('--size', '-s', type=click.INT, nargs=3,
              default=(64, 64, 64), help='the size of created chunk')
@click.option('--dtype', '-d',
              type=click.Choice(['uint8', 'uint32', 'uint16', 'uint64',
                                 'float32', 'float64']),
              default='uint8', help='the data type of chunk')
@click.option('--pattern', '-p', type=click.Choice(['sin', 'zero', 'random']),
              default='sin', help='ways to generate array.')
@click.option('--voxel-offset', '-t', type=click.INT, nargs=3,
              default=(0, 0, 0), help='offset in voxel number.')
@click.option('--voxel-size', '-e', type=click.INT, nargs=3,
              default=(1, 1, 1), help='voxel size in nm')
@click.option('--output-chunk-name', '-o', type=str, default='chunk',
              help='name of created chunk')
@operator
def create_chunk(tasks, size, dtype, pattern, voxel_offset, voxel_size,
                 output_chunk_name):
    """Create a fake chunk for easy test."""
    print(f'creating chunk: {output_chunk_name}')
    for task in tasks:
        if task is not None:
            if 'bbox' in task:
                task[output_chunk_name] = Chunk.from_bbox(
                    task['bbox'], dtype=np.dtype(dtype), pattern=pattern,
                    voxel_size=voxel_size)
            else:
                task[output_chunk_name] = Chunk.create(
                    size=size, dtype=np.dtype(dtype), pattern=pattern,
                    voxel_offset=voxel_offset, voxel_size=voxel_size)
        yield task


@main.command('normalize-intensity')
@click.option('--name', type=str, default='normalize-intensity',
              help='name of operator')
@click.option('--input-chunk-name', '-i', type=str,
              default=DEFAULT_CHUNK_NAME, help='input chunk name')
@click.option('--output-chunk-name', '-o', type=str,
              default=DEFAULT_CHUNK_NAME, help='output chunk name')
@operator
def normalize_intensity(tasks, name, input_chunk_name, output_chunk_name):
    """transform gray image to float (-1:1). x=(x-127.5) - 1.0"""
    import torch
    ops = None
    for task in tasks:
        if task is not None:
            start = time()
            chunk = task[input_chunk_name]
            if torch.cuda.is_available():
                # device path: upload once, HIP kernel (flow.py:1664-1666)
                from .ops import HipOps
                if ops is None:
                    ops = HipOps(0)
                chunk = chunk.to_device()
                assert chunk.array.dtype == torch.uint8
                out = ops.normalize_intensity(chunk.array)
            else:
                arr = chunk.numpy().array
                assert np.issubdtype(arr.dtype, np.uint8)
                out = arr.astype('float32')
                out /= 127.5
                out -= 1.0
            task[output_chunk_name] = Chunk(
                out, voxel_offset=chunk.voxel_offset,
                voxel_size=chunk.voxel_size)
            task['log']['timer'][name] = time() - start
        yield task


@main.command('normalize-contrast')
@click.option('--name', type=str, default='normalize-contrast-nkem',
              help='name of operator.')
@click.option('--input-chunk-name', '-i', type=str,
              default=DEFAULT_CHUNK_NAME, help='input chunk name')
@click.option('--output-chunk-name', '-o', type=str,
              default=DEFAULT_CHUNK_NAME, help='output chunk name')
@click.option('--lower-clip-fraction', '-l', type=click.FLOAT, default=0.01,
              help='lower intensity fraction to clip out.')
@click.option('--upper-clip-fraction', '-u', type=click.FLOAT, default=0.01,
              help='upper intensity fraction to clip out.')
@click.option('--minval', type=click.INT, default=1,
              help='the minimum intensity of transformed chunk.')
@click.option('--maxval', type=click.INT, default=255,
              help='the maximum intensity of transformed chunk.')
@click.option('--per-section/--whole', default=True,
              help='per section normalization or normalize the whole chunk.')
@operator
def normalize_contrast(tasks, name, input_chunk_name, output_chunk_name,
                       lower_clip_fraction, upper_clip_fraction, minval,
                       maxval, per_section):
    """Normalize the section contrast using precomputed histograms."""
    import torch
    from .contrast import normalize_contrast as _nc
    for task in tasks:
        if task is not None:
            start = time()
            chunk = task[input_chunk_name]
            if torch.cuda.is_available():
                chunk = chunk.to_device().clone()
            else:
                chunk = chunk.clone()
            task[output_chunk_name] = _nc(
                chunk, lower_clip_fraction=lower_clip_fraction,
                upper_clip_fraction=upper_clip_fraction, minval=minval,
                maxval=maxval, per_section=per_section)
            task['log']['timer'][name] = time() - start
        yield task


@main.command('inference')
@click.option('--name', type=str, default='inference',
              help='name of this operator')
@click.option('--convnet-model', '-m', type=str, default=None,
              help='convnet model path or type.')
@click.option('--convnet-weight-path', '-w', type=str, default=None,
              help='convnet weight path')
@click.option('--input-patch-size', '-s', type=click.INT, nargs=3,
              required=True, help='input patch size')
@click.option('--output-patch-size', '-z', type=click.INT, nargs=3,
              default=None, callback=default_none, help='output patch size')
@click.option('--output-patch-overlap', '-v', type=click.INT, nargs=3,
              default=(4, 64, 64), help='patch overlap')
@click.option('--output-crop-margin', type=click.INT, nargs=3, default=None,
              callback=default_none, help='margin size of output chunk cropping.')
@click.option('--patch-num', '-n', default=None, callback=default_none,
              type=click.INT, nargs=3, help='patch number in z,y,x.')
@click.option('--num-input-channels', type=click.INT, default=1,
              help='number of input channels')
@click.option('--num-output-channels', '-c', type=click.INT, default=3,
              help='number of output channels')
@click.option('--dtype', '-d',
              type=click.Choice(['float32', 'float16', 'bfloat16']),
              default='float32',
              help='output data type. bfloat16 is an MI355X extension of '
                   'the reference choice (flow.py:1871-1875): compute runs '
                   'in bf16 on the MFMA cores while blending stays f32, so '
                   'the output chunk is float32 (numpy has no bfloat16).')
@click.option('--framework', '-f',
              type=click.Choice(['universal', 'identity', 'pytorch']),
              default='universal', help='inference framework')
@click.option('--batch-size', '-b', type=click.INT, default=1,
              help='mini batch size of input patch.')
@click.option('--bump', type=click.Choice(['wu', 'zung']), default='wu',
              help='bump function type (only support wu now!).')
@click.option('--mask-output-chunk/--no-mask-output-chunk', default=False,
              help='mask output chunk will make the whole chunk like one '
                   'output patch. This will also work with non-aligned chunk size.')
@click.option('--mask-myelin-threshold', '-y', default=None, type=click.FLOAT,
              help='mask myelin if netoutput have myelin channel.')
@click.option('--augment/--no-augment', default=False,
              help='transform the input patch and transform back the output patch')
@click.option('--input-chunk-name', '-i', type=str, default='chunk',
              help='input chunk name')
@click.option('--output-chunk-name', '-o', type=str, default='chunk',
              help='output chunk name')
@operator
def inference(tasks, name, convnet_model, convnet_weight_path,
              input_patch_size, output_patch_size, output_patch_overlap,
              output_crop_margin, patch_num, num_input_channels,
              num_output_channels, dtype, framework, batch_size, bump,
              mask_output_chunk, mask_myelin_threshold, augment,
              input_chunk_name, output_chunk_name):
    """Perform convolutional network inference for chunks."""
    from .inferencer import Inferencer
    with Inferencer(
            convnet_model, convnet_weight_path,
            input_patch_size=input_patch_size,
            output_patch_size=output_patch_size,
            num_input_channels=num_input_channels,
            num_output_channels=num_output_channels,
            output_patch_overlap=output_patch_overlap,
            output_crop_margin=output_crop_margin,
            patch_num=patch_num, framework=framework, dtype=dtype,
            batch_size=batch_size, bump=bump, augment=augment,
            mask_output_chunk=mask_output_chunk,
            mask_myelin_threshold=mask_myelin_threshold,
            dry_run=state['dry_run']) as inferencer:
        for task in tasks:
            if task is not None:
                if 'log' not in task:
                    task['log'] = {'timer': {}}
                start = time()
                task[output_chunk_name] = inferencer(task[input_chunk_name])
                task['log']['timer'][name] = time() - start
                task['log']['compute_device'] = inferencer.compute_device
            yield task


@main.command('crop-margin')
@click.option('--name', type=str, default='crop-margin',
              help='name of this operator')
@click.option('--margin-size', '-m', type=click.INT, nargs=6, default=None,
              callback=default_none,
              help='crop the chunk margin. It should have 6 values.')
@click.option('--crop-bbox/--no-crop-bbox', default=False,
              help='adjust the bounding box or not.')
@click.option('--input-chunk-name', '-i', type=str, default='chunk',
              help='input chunk name.')
@click.option('--output-chunk-name', '-o', type=str, default='chunk',
              help='output chunk name.')
@operator
def crop_margin(tasks, name, margin_size, crop_bbox, input_chunk_name,
                output_chunk_name):
    """Crop the margin of chunk."""
    ops = None
    for task in tasks:
        if task is not None:
            start = time()
            chunk = task[input_chunk_name]
            if margin_size:
                if chunk.is_device:
                    # HIP contiguous-copy kernel (chunk/base.py:691-726)
                    from .ops import HipOps
                    if ops is None:
                        ops = HipOps(0)
                    out = ops.crop_margin(chunk.array, list(margin_size))
                    offset = tuple(o + m for o, m in
                                   zip(chunk.voxel_offset, margin_size[:3]))
                    task[output_chunk_name] = Chunk(
                        out, voxel_offset=offset,
                        voxel_size=chunk.voxel_size)
                else:
                    task[output_chunk_name] = chunk.crop_margin(
                        margin_size=margin_size)
            else:
                task[output_chunk_name] = chunk.cutout(task['bbox'].slices)
            task['log']['timer'][name] = time() - start
        yield task


@main.command('connected-components')
@click.option('--name', type=str, default='connected-components',
              help='threshold a map and get the targets.')
@click.option('--input-chunk-name', '-i', type=str,
              default=DEFAULT_CHUNK_NAME, help='input chunk name')
@click.option('--output-chunk-name', '-o', type=str,
              default=DEFAULT_CHUNK_NAME, help='output chunk name')
@click.option('--threshold', '-t', type=click.FLOAT, default=None,
              help='threshold to cut the map.')
@click.option('--connectivity', '-c', type=click.Choice(['6', '18', '26']),
              default='6', help='number of neighboring voxels used.')
@operator
def connected_components(tasks, name, input_chunk_name, output_chunk_name,
                         threshold, connectivity):
    """Threshold the probability map to get a segmentation."""
    from .connected import connected_component
    connectivity = int(connectivity)
    for task in tasks:
        if task is not None:
            start = time()
            task[output_chunk_name] = connected_component(
                task[input_chunk_name], threshold=threshold,
                connectivity=connectivity)
            task['log']['timer'][name] = time() - start
        yield task


@main.command('plugin')
@click.option('--name', type=str, default='plugin-1',
              help='name of plugin. Multiple plugins should have different names.')
@click.option('--input-names', '-i', type=str, default=None,
              help='input names with delimiter of comma')
@click.option('--output-names', '-o', type=str, default=None,
              help='output names with dilimiter of comma')
@click.option('--file', '-f', type=str, help='python file to call.')
@click.option('--args', '-a', type=str, default=None,
              help='arguments of plugin.')
@operator
def plugin(tasks, name, input_names, output_names, file, args):
    """Insert custom program as a plugin."""
    op = Plugin(file, name=name)
    for task in tasks:
        if task is not None:
            start = time()
            if input_names is not None:
                inputs = [None if n == 'None' else task[n]
                          for n in input_names.split(',')]
            else:
                inputs = []
            outputs = op(inputs, args=args)
            if isinstance(outputs, (list, tuple)):
                names = output_names.split(',')
                assert len(outputs) == len(names)
                for n, o in zip(names, outputs):
                    task[n] = o
            elif output_names is not None:
                assert ',' not in output_names
                task[output_names] = outputs
            else:
                assert outputs is None
            task['log']['timer'][name] = time() - start
        yield task


@main.command('save-npy')
@click.option('--name', type=str, default='save-npy', help='name of operator')
@click.option('--file-name', '-f', type=str, required=True,
              help='output .npy path; {bbox} expands to the task bbox string')
@click.option('--input-chunk-name', '-i', type=str,
              default=DEFAULT_CHUNK_NAME, help='input chunk name')
@operator
def save_npy(tasks, name, file_name, input_chunk_name):
    """Save a chunk as .npy (D2H at the pipeline edge)."""
    for task in tasks:
        if task is not None:
            start = time()
            fname = file_name
            if '{bbox}' in fname and 'bbox' in task:
                fname = fname.replace('{bbox}', task['bbox'].string)
            task[input_chunk_name].to_npy(fname)
            task['log']['timer'][name] = time() - start
        yield task


@main.command('stitch')
@click.option('--name', type=str, default='stitch', help='name of operator')
@click.option('--input-chunk-name', '-i', type=str,
              default=DEFAULT_CHUNK_NAME, help='input chunk name')
@click.option('--output-volume-name', '-o', type=str, default='volume',
              help='name of the stitched volume in the final task')
@click.option('--file-name', '-f', type=str, default=None,
              help='save the stitched volume as .npy on rank 0')
@click.option('--backend', type=str, default=None,
              help='torch.distributed backend override (tests use gloo; '
                   'default nccl=RCCL on GPU, gloo otherwise)')
@operator
def stitch(tasks, name, input_chunk_name, output_volume_name, file_name,
           backend):
    """Gather per-rank output chunks into one volume on rank 0.

    The terminal operator of the multi-GPU worker mode (BASELINE config 3):
    each torchrun rank runs the same pipeline over its task shard
    (generate-tasks --task-rank/--task-world); this op collects the rank's
    output chunks, then performs the single RCCL-over-xGMI collective of
    the whole pipeline — a p2p gather of disjoint sub-volumes to rank 0
    (dispatch.stitch_to_rank0, the same code bench.py times). Rank 0 yields
    one final task carrying the stitched volume; other ranks yield nothing.
    """
    import torch
    from .dispatch import init_distributed, stitch_to_rank0
    collected = {}
    last_task = None
    for task in tasks:
        if task is not None:
            start = time()
            chunk = task[input_chunk_name]
            t = chunk.array
            if not isinstance(t, torch.Tensor):
                t = torch.from_numpy(np.ascontiguousarray(t))
            if t.dtype != torch.float32:
                t = t.to(torch.float32)
            if t.ndim == 3:
                t = t[None]
            idx = task.get('bbox_index', len(collected))
            collected[idx] = (task.get('bbox'), t)
            last_task = task
            last_task['log']['timer'][name] = time() - start
    if collected and any(b is None for b, _ in collected.values()):
        raise ValueError('stitch needs bbox-carrying tasks — put '
                         'generate-tasks at the head of the pipeline')
    rank, world = init_distributed(backend=backend)
    all_bboxes = state.get('all_bboxes')
    if all_bboxes is None:
        assert world == 1, \
            'stitch with world_size>1 needs generate-tasks in the pipeline'
        all_bboxes = [b for b, _ in
                      (collected[i] for i in sorted(collected))]
    if not collected:
        return
    channels = next(iter(collected.values()))[1].shape[0]
    device = next(iter(collected.values()))[1].device
    local = {i: t for i, (_, t) in collected.items()}
    volume = stitch_to_rank0(all_bboxes, local, channels, rank, world,
                             device)
    if rank == 0:
        roi_start = tuple(min(b.start[d] for b in all_bboxes)
                          for d in range(3))
        vol_chunk = Chunk(volume, voxel_offset=roi_start)
        if file_name:
            vol_chunk.to_npy(file_name)
        task = last_task or get_initial_task()
        task[output_volume_name] = vol_chunk
        yield task


@main.command('load-npy')
@click.option('--name', type=str, default='load-npy', help='name of operator')
@click.option('--file-name', '-f', type=str, required=True,
              help='input .npy path')
@click.option('--voxel-offset', '-t', type=click.INT, nargs=3,
              default=(0, 0, 0), help='voxel offset')
@click.option('--output-chunk-name', '-o', type=str,
              default=DEFAULT_CHUNK_NAME, help='output chunk name')
@operator
def load_npy(tasks, name, file_name, voxel_offset, output_chunk_name):
    """Load a chunk from .npy."""
    for task in tasks:
        if task is not None:
            start = time()
            task[output_chunk_name] = Chunk.from_npy(
                file_name, voxel_offset=voxel_offset)
            task['log']['timer'][name] = time() - start
        yield task


@main.command('load-h5')
@click.option('--name', type=str, default='load-h5', help='name of operator')
@click.option('--file-name', '-f', type=str, required=True,
              help='HDF5 file, or a per-bbox file prefix')
@click.option('--dataset-path', '-d', type=str, default=None,
              help='dataset path inside the HDF5 file')
@click.option('--dtype', '-e', default=None,
              type=click.Choice(['float32', 'float64', 'uint16', 'uint32',
                                 'uint64', 'uint8']),
              help='transform data type')
@click.option('--voxel-offset', '-v', type=click.INT, nargs=3, default=None,
              callback=default_none, help='voxel offset of the dataset')
@click.option('--voxel-size', '-x', type=click.INT, nargs=3, default=None,
              callback=default_none, help='physical voxel size (nm)')
@click.option('--cutout-start', '-t', type=click.INT, nargs=3,
              callback=default_none, help='cutout start (global)')
@click.option('--cutout-stop', '-p', type=click.INT, nargs=3,
              callback=default_none, help='cutout stop (global)')
@click.option('--cutout-size', '-s', type=click.INT, nargs=3,
              callback=default_none, help='cutout size')
@click.option('--set-bbox/--no-set-bbox', default=False,
              help='set the task bbox from the loaded chunk')
@click.option('--output-chunk-name', '-o', type=str,
              default=DEFAULT_CHUNK_NAME, help='output chunk name')
@operator
def load_h5(tasks, name, file_name, dataset_path, dtype, voxel_offset,
            voxel_size, cutout_start, cutout_stop, cutout_size, set_bbox,
            output_chunk_name):
    """Read HDF5 files (reference flow.py:976-1065; in-repo codec —
    a task bbox supplies the cutout and per-bbox file name)."""
    for task in tasks:
        if task is not None:
            start = time()
            cs, cp = cutout_start, cutout_stop
            csz = cutout_size
            fname = file_name
            if 'bbox' in task and cutout_start is None:
                bbox = task['bbox']
                cs, cp = tuple(bbox.start), tuple(bbox.stop)
                csz = tuple(bbox.shape)
                if not file_name.endswith('.h5'):
                    fname = f'{file_name}{bbox.string}.h5'
            chunk = Chunk.from_h5(
                fname, dataset_path=dataset_path,
                voxel_offset=voxel_offset, voxel_size=voxel_size,
                cutout_start=cs, cutout_stop=cp, cutout_size=csz,
                dtype=dtype)
            if chunk is not None and dtype is not None:
                chunk = chunk.astype(dtype)
            task[output_chunk_name] = chunk
            if set_bbox and chunk is not None:
                task['bbox'] = chunk.bbox
            task['log']['timer'][name] = time() - start
        yield task


@main.command('save-h5')
@click.option('--name', type=str, default='save-h5', help='name of operator')
@click.option('--input-name', '-i', type=str, default=DEFAULT_CHUNK_NAME,
              help='input chunk name')
@click.option('--file-name', '-f', type=str, required=True,
              help='file name or prefix of the output HDF5 file')
@click.option('--with-offset/--without-offset', default=True,
              help='store the /voxel_offset dataset or not')
@click.option('--voxel-size', '-v', type=click.INT, nargs=3, default=None,
              callback=default_none, help='voxel size to store')
@click.option('--dtype', '-d', type=str, default=None,
              help='data type conversion before saving')
@operator
def save_h5(tasks, name, input_name, file_name, with_offset, voxel_size,
            dtype):
    """Save chunk to an HDF5 file (reference flow.py:1068-1120)."""
    for task in tasks:
        if task is not None:
            start = time()
            chunk = task[input_name]
            if dtype is not None:
                chunk = chunk.astype(dtype)
            chunk.to_h5(file_name, with_offset=with_offset,
                        voxel_size=voxel_size)
            task['log']['timer'][name] = time() - start
        yield task


@main.command('load-tif')
@click.option('--name', type=str, default='load-tif', help='name of operator')
@click.option('--file-name', '-f', required=True,
              type=click.Path(exists=True, dir_okay=True),
              help='TIFF file, or a directory of per-section *.tif* files')
@click.option('--voxel-offset', '-v', type=click.INT, nargs=3, default=None,
              callback=default_none, help='global offset of this chunk')
@click.option('--voxel-size', '-s', type=click.INT, nargs=3, default=None,
              callback=default_none, help='physical voxel size (nm)')
@click.option('--dtype', '-d', default=None,
              type=click.Choice(['uint8', 'uint16', 'uint32', 'uint64',
                                 'float32', 'float64', 'float16']),
              help='convert to data type')
@click.option('--output-chunk-name', '-o', type=str,
              default=DEFAULT_CHUNK_NAME, help='output chunk name')
@operator
def load_tif(tasks, name, file_name, voxel_offset, voxel_size, dtype,
             output_chunk_name):
    """Read TIFF files (reference flow.py:918-950; in-repo codec)."""
    for task in tasks:
        if task is not None:
            start = time()
            task[output_chunk_name] = Chunk.from_tif(
                file_name, dtype=dtype, voxel_offset=voxel_offset,
                voxel_size=voxel_size)
            task['log']['timer'][name] = time() - start
        yield task


@main.command('save-tif')
@click.option('--name', type=str, default='save-tif', help='name of operator')
@click.option('--input-chunk-name', '-i', type=str,
              default=DEFAULT_CHUNK_NAME, help='input chunk name')
@click.option('--file-name', '-f', default=None, type=str,
              help='output .tif/.tiff path; default <bbox>.tif')
@click.option('--dtype', '-t', default=None,
              type=click.Choice(['uint8', 'uint16', 'uint32', 'uint64',
                                 'float32', 'float64']),
              help='convert to this data type before saving')
@click.option('--compression', '-c', type=click.Choice(['', 'zlib']),
              default='zlib',
              help='codec (in-repo writer: raw or zlib/deflate)')
@operator
def save_tif(tasks, name, input_chunk_name, file_name, dtype, compression):
    """Save chunk as a TIFF file (reference flow.py:953-974)."""
    for task in tasks:
        if task is not None:
            start = time()
            chunk = task[input_chunk_name]
            if dtype:
                chunk = chunk.astype(dtype)
            chunk.to_tif(file_name, compression=compression)
            task['log']['timer'][name] = time() - start
        yield task


@main.command('save-log')
@click.option('--output-path', '-o', type=str, required=True,
              help='directory for per-task log JSON files')
@operator
def save_log(tasks, output_path):
    """Persist each task's timer log as JSON (reference upload_log shape)."""
    os.makedirs(output_path, exist_ok=True)
    for idx, task in enumerate(tasks):
        if task is not None:
            tag = task.get('bbox').string if 'bbox' in task else str(idx)
            with open(os.path.join(output_path, f'{tag}.json'), 'w') as f:
                json.dump(task['log'], f)
        yield task


@main.command('log-summary')
@click.option('--log-dir', '-l', type=click.Path(exists=True, dir_okay=True),
              required=True, help='directory of json log files.')
@click.option('--output-size', '-s', type=click.INT, nargs=3, default=None,
              callback=default_none, help='output size for voxels/sec')
@generator
def log_summary(log_dir, output_size):
    """Compute the statistics of large scale run (the reference's
    per-operator mean/max/min + voxels/sec report, log_summary.py:16-74)."""
    import glob
    timers = {}
    for fname in glob.glob(os.path.join(log_dir, '*.json')):
        with open(fname) as f:
            log = json.load(f)
        for op_name, secs in log.get('timer', {}).items():
            timers.setdefault(op_name, []).append(secs)
    total_mean = 0.0
    for op_name, vals in sorted(timers.items()):
        arr = np.array(vals)
        print(f'{op_name}: mean={arr.mean():.3f}s max={arr.max():.3f}s '
              f'min={arr.min():.3f}s n={len(arr)}')
        total_mean += arr.mean()
    if output_size is not None and total_mean > 0:
        voxels = int(np.prod(output_size))
        print(f'speed: {voxels / total_mean / 1e3:.1f} kv/s '
              f'({voxels / total_mean / 1e6:.2f} mv/s)')
    yield get_initial_task()


@main.command('copy-var')
@click.option('--from-name', '-f', type=str, default='chunk',
              help='Variable to be copied')
@click.option('--to-name', '-t', type=str, default='chunk',
              help='New variable name')
@click.option('--deep-copy/--shallow-copy', type=bool, default=True,
              help='really copy data or just create a new name or reference.')
@operator
def copy_var(tasks, from_name, to_name, deep_copy):
    """Deep or shallow copy a variable."""
    from copy import deepcopy
    for task in tasks:
        if task is not None:
            if deep_copy:
                v = task[from_name]
                task[to_name] = v.clone() if isinstance(v, Chunk) \
                    else deepcopy(v)
            else:
                task[to_name] = task[from_name]
        yield task


@main.command('delete-var')
@click.option('--var-names', '-v', type=str, required=True,
              help='the variable names to be deleted')
@operator
def delete_var(tasks, var_names):
    """Delete a Chunk in task to release RAM (or HBM)."""
    for task in tasks:
        if task is not None:
            for var_name in var_names.split(','):
                del task[var_name]
        yield task


@main.command('skip-task-by-file')
@click.option('--prefix', '-p', required=True, type=str,
              help='the pre part of result file path')
@click.option('--suffix', '-s', default='', type=str,
              help='the post part of result file path.')
@click.option('--mode', '-m',
              type=click.Choice(['missing', 'empty', 'exist']),
              default='exist',
              help='skip if the corresponding file is missing/empty/exists')
@click.option('--adjust-size', '-a', default=None, type=click.INT,
              callback=default_none, help='expand or shrink the bounding box.')
@operator
def skip_task_by_file(tasks, prefix, suffix, mode, adjust_size):
    """if a result file already exists, skip this task (the reference's
    at-least-once idempotency family — SURVEY.md §5)."""
    for task in tasks:
        if task is not None:
            bbox = task['bbox']
            if adjust_size is not None:
                bbox = bbox.adjust(adjust_size)
            fname = prefix + bbox.string + suffix
            if mode == 'empty':
                if not os.path.exists(fname) or os.path.getsize(fname) == 0:
                    task = None
            elif mode == 'missing':
                if not os.path.exists(fname):
                    task = None
            elif mode == 'exist':
                if os.path.exists(fname):
                    task = None
        yield task


@main.command('mark-complete')
@click.option('--prefix', '-p', type=str, required=True,
              help='pre-path of a file.')
@click.option('--suffix', '-s', type=str, default='',
              help='suffix of the flag file.')
@operator
def mark_complete(tasks, prefix, suffix):
    """mark completion of a task as an empty file."""
    from pathlib import Path
    for task in tasks:
        if task is not None:
            Path(f"{prefix}{task['bbox'].string}{suffix}").touch()
        yield task


@main.command('skip-all-zero')
@click.option('--input-chunk-name', '-i', type=str,
              default=DEFAULT_CHUNK_NAME, help='input chunk name')
@click.option('--prefix', '-p', type=str, default=None,
              help='pre-path of a trace file.')
@click.option('--suffix', '-s', type=str, default='',
              help='post-path of a trace file.')
@click.option('--adjust-size', '-a', type=click.INT, default=None,
              help='change the bounding box of the trace file name.')
@click.option('--chunk-bbox/--task-bbox', default=True,
              help='use the chunk bounding box or the task one.')
@operator
def skip_all_zero(tasks, input_chunk_name, prefix, suffix, adjust_size,
                  chunk_bbox):
    """if chunk has all zero, skip this task."""
    import torch
    from pathlib import Path
    for task in tasks:
        if task is not None:
            chunk = task[input_chunk_name]
            arr = chunk.array
            any_nonzero = bool((arr != 0).any().item()) \
                if isinstance(arr, torch.Tensor) else bool(np.any(arr))
            if not any_nonzero:
                if prefix is not None:
                    bbox = chunk.bbox if chunk_bbox else task['bbox']
                    if adjust_size is not None:
                        bbox = bbox.adjust(adjust_size)
                    fname = f'{prefix}{bbox.string}{suffix}'
                    if not os.path.exists(fname):
                        Path(fname).touch()
                task = None
        yield task


@main.command('skip-none')
@click.option('--input-name', '-i', type=str, default=DEFAULT_CHUNK_NAME,
              help='input name')
@click.option('--touch/--no-touch', default=True,
              help='touch an empty file or not')
@click.option('--prefix', '-p', default=None, help='prefix of output file.')
@click.option('--suffix', '-s', default=None, help='suffix of output file.')
@operator
def skip_none(tasks, input_name, touch, prefix, suffix):
    """If item is None, skip this task."""
    from pathlib import Path
    for task in tasks:
        if task is not None:
            if task[input_name] is None:
                if touch:
                    assert prefix is not None and suffix is not None
                    Path(f"{prefix}{task['bbox'].string}{suffix}").touch()
                task = None
        yield task


if __name__ == '__main__':  # python -m chunkflow_amd.flow <ops...>
    main()
