"""train_and_validate_federated at bench-class config: --hip_graph on/off
wall time per local step (VERDICT r1 #7 'Done' check)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault('FEDTORCH_SYNTH_SIZE', '4096')
import torch
from fedtorch_amd.parameters import get_args
from fedtorch_amd.nodes import Client
from fedtorch_amd.trainings.federated import train_and_validate_federated

def run(hg):
    argv = ['-d', 'cifar10', '-a', 'resnet20', '-f', 'true',
            '--federated_type', 'fedavg', '--num_comms', '12',
            '--online_client_rate', '1.0',
            '--federated_sync_type', 'local_step', '--local_step', '10',
            '-b', '256', '--lr', '0.1', '--in_momentum', 'true',
            '--weight_decay', '5e-4', '--on_cuda', 'true', '--bf16', 'true',
            '--channels_last', 'true', '--fused_bn', 'true',
            '--hip_graph', 'true' if hg else 'false',
            '--debug', 'false', '-j', '0', '--manual_seed', '7',
            '--checkpoint', '/tmp/ft_lgs_%d' % int(hg)]
    args = get_args(argv)
    c = Client(args, 0)
    c.initialize(); c.initialize_dataset(); c.load_local_dataset()
    c.gen_aux_models()
    # warm (find + capture) untimed
    c.args.num_comms = 3
    train_and_validate_federated(c, validate=False)
    torch.cuda.synchronize()
    s0 = c.args.local_index
    c.args.num_comms = 10
    t0 = time.perf_counter()
    train_and_validate_federated(c, validate=False)
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    steps = c.args.local_index - s0
    print('hip_graph=%d: %.2f ms/local-step (%d steps, %.2fs total)'
          % (hg, el / max(steps, 1) * 1e3, steps, el), flush=True)
    return el / max(steps, 1)

t_on = run(True)
t_off = run(False)
print('speedup: %.2fx' % (t_off / t_on), flush=True)
