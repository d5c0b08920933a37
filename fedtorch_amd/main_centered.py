# -*- coding: utf-8 -*-
"""Centered entry (parity with reference `main_centered.py`): the whole
federation simulated in ONE process on ONE GPU — N virtual clients whose
replicas and aux state stay resident in the MI355X's 288 GB HBM3E."""
from fedtorch_amd.parameters import get_args
from fedtorch_amd.nodes.centered import ClientCentered, ServerCentered
from fedtorch_amd.trainings.centered.main import (
    train_and_validate_federated_centered)


def main(args):
    ClientNodes = {}
    for i in range(args.num_workers):
        if args.data in ('emnist', 'emnist_full', 'synthetic') or i == 0:
            ClientNodes[i] = ClientCentered(args, i)
        else:
            ClientNodes[i] = ClientCentered(
                args, i, Partitioner=ClientNodes[0].Partitioner)
    ServerNode = ServerCentered(ClientNodes[0].args, ClientNodes[0].model)
    ServerNode.enable_grad(ClientNodes[0].train_loader)

    if ServerNode.args.federated_drfa:
        from fedtorch_amd.trainings.centered.drfa import \
            train_and_validate_drfa_centered
        train_and_validate_drfa_centered(ClientNodes, ServerNode)
    else:
        t = ServerNode.args.federated_type
        if t == 'apfl':
            from fedtorch_amd.trainings.centered.apfl import \
                train_and_validate_apfl_centered
            train_and_validate_apfl_centered(ClientNodes, ServerNode)
        elif t == 'perfedme':
            from fedtorch_amd.trainings.centered.perfedme import \
                train_and_validate_perfedme_centered
            train_and_validate_perfedme_centered(ClientNodes, ServerNode)
        elif t == 'afl':
            from fedtorch_amd.trainings.centered.afl import \
                train_and_validate_afl_centered
            train_and_validate_afl_centered(ClientNodes, ServerNode)
        elif t in ('fedavg', 'scaffold', 'fedgate', 'qsparse', 'fedprox',
                   'qffl', 'perfedavg', 'fedadam'):
            train_and_validate_federated_centered(ClientNodes, ServerNode)
        else:
            raise NotImplementedError(t)


if __name__ == '__main__':
    main(get_args())
