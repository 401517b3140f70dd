"""Elementwise-tail attribution: torch profiler with input shapes over
one graphless train step at bench config, printing the top CUDA-time
ops with shapes (the kernel-stats CSV can't attribute elementwise
launches to python sites; shapes narrow them down)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..'))
import torch
from torch.profiler import profile, ProfilerActivity

from lingvo_amd.core import registry

model_p = registry.GetParams('asr.librispeech.Librispeech960WpmConformerL',
                             'Train')
model_p.input.batch_size = int(sys.argv[1]) if len(sys.argv) > 1 else 512
model_p.task.random_seed = 1234
task = model_p.Instantiate().to('cuda').GetTask()
batch = task.input_generator.GetPreprocessedInputBatch().Transform(
    lambda t: t.to('cuda') if isinstance(t, torch.Tensor) else t)
for _ in range(2):
  task.TrainStep(batch)
torch.cuda.synchronize()
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True) as prof:
  task.TrainStep(batch)
  torch.cuda.synchronize()
print(prof.key_averages(group_by_input_shape=True).table(
    sort_by='cuda_time_total', row_limit=45, max_src_column_width=60))
