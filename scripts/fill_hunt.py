"""Attribute the per-step FillFunctor/zero launches (one eager flagship
step, torch profiler with stacks)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from fedtorch_amd.parameters import get_args
from fedtorch_amd.nodes import Client
from fedtorch_amd.trainings.federated import amp

args = get_args(['-d','cifar10','-a','resnet20','-f','true',
 '--federated_type','fedavg','--num_comms','1','-b','256','--lr','0.1',
 '--bf16','true','--channels_last','true','--in_momentum','true',
 '-j','0','--checkpoint','/tmp/fh','--debug','false'])
os.environ.setdefault('FEDTORCH_SYNTH_SIZE','2048')
client = Client(args, 0); client.initialize(); client.gen_aux_models()
args = client.args
x = torch.randn(256,3,32,32,device='cuda').to(memory_format=torch.channels_last)
y = torch.randint(0,10,(256,),device='cuda')
def step():
    client.optimizer.zero_grad()
    with amp(args):
        loss = client.criterion(client.model(x), y)
    loss.backward()
    client.optimizer.step(apply_lr=True, apply_in_momentum=True,
                          apply_out_momentum=False)
for _ in range(6):
    step()
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True) as prof:
    step()
    torch.cuda.synchronize()
for e in prof.key_averages(group_by_input_shape=True):
    if 'fill' in e.key.lower() or 'zero' in e.key.lower():
        print(e.key, 'count', e.count, 'shapes', e.input_shapes,
              'cuda_us', getattr(e, 'self_device_time_total', 0))
