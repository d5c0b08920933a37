set -x
mkdir -p gpurun_out
# quantized fedavg on GPU (exercises quantize/dequant-accumulate kernels in situ)
timeout 300 env FEDTORCH_SYNTH_SIZE=2048 python -c "
from fedtorch_amd.parameters import get_args
from fedtorch_amd.main import main
main(get_args(['-d','cifar10','-a','resnet20','-f','true','--federated_type','fedavg',
 '--quantized','true','--num_comms','2','--online_client_rate','1.0','-b','64','--lr','0.1',
 '--bf16','true','-j','0','--checkpoint','/tmp/ckq','--debug','false']))
print('QUANT-GPU OK')" > gpurun_out/quant_gpu.log 2>&1; echo QUANT=$?
# compressed fedgate on GPU (topk + scatter-accumulate kernels in situ)
timeout 300 env FEDTORCH_SYNTH_SIZE=2048 python -c "
from fedtorch_amd.parameters import get_args
from fedtorch_amd.main import main
main(get_args(['-d','cifar10','-a','resnet20','-f','true','--federated_type','fedgate',
 '--compressed','true','--compressed_ratio','0.2','--num_comms','2','--online_client_rate','1.0',
 '-b','64','--lr','0.1','--bf16','true','-j','0','--checkpoint','/tmp/ckc','--debug','false']))
print('COMP-GPU OK')" > gpurun_out/comp_gpu.log 2>&1; echo COMP=$?
# packed 16 virtual clients on GPU
timeout 300 env FEDTORCH_SYNTH_SIZE=4096 python -c "
from fedtorch_amd.parameters import get_args
from fedtorch_amd.main import main
main(get_args(['-d','cifar10','-a','resnet20','-f','true','--federated_type','fedavg',
 '--num_comms','2','--online_client_rate','0.5','--local_step','4','--federated_sync_type','local_step',
 '--clients_per_rank','16','--in_momentum','true','-b','64','--lr','0.1','--bf16','true','-j','0',
 '--checkpoint','/tmp/ckp','--debug','false']))
print('PACKED-GPU OK')" > gpurun_out/packed_gpu.log 2>&1; echo PACKED=$?
# centered 8 clients on GPU (scaffold)
timeout 300 env FEDTORCH_SYNTH_SIZE=2048 python -c "
from fedtorch_amd.parameters import get_args
from fedtorch_amd.main_centered import main
main(get_args(['-d','cifar10','-a','resnet20','-f','true','--federated_type','scaffold',
 '--num_comms','2','--online_client_rate','0.5','--num_workers','8','-b','64','--lr','0.1',
 '--bf16','true','--checkpoint','/tmp/ckcen','--debug','false']))
print('CENTERED-GPU OK')" > gpurun_out/centered_gpu.log 2>&1; echo CENTERED=$?
for f in gpurun_out/quant_gpu.log gpurun_out/comp_gpu.log gpurun_out/packed_gpu.log gpurun_out/centered_gpu.log; do tail -1 $f; done
