# -*- coding: utf-8 -*-
"""Console/record logging (parity with reference `fedtorch/logs/logging.py`).

Keeps the `record{rank}` per-rank log file format that the offline tools
(`fedtorch_amd/tools/load_console_records.py`) regex-parse.
"""
import os
import time

from fedtorch_amd.utils.op_files import write_txt

log_path = None


def configure_log(args=None):
    global log_path
    if args is not None:
        log_path = os.path.join(
            args.checkpoint_dir, 'record' + str(args.graph.rank))
    else:
        log_path = os.path.join(os.getcwd(), 'record')


def log(content, debug=True):
    """Print + append to the per-rank record file (reference `logging.py:26-31`)."""
    content = time.strftime('%Y:%m:%d %H:%M:%S') + '\t' + content
    if debug:
        print(content, flush=True)
        if log_path is not None:
            try:
                write_txt(content + '\n', log_path, type='a')
            except OSError:
                pass


def log_args(args, debug=True):
    log('parameters: ', debug=debug)
    for arg in vars(args):
        log(str(arg) + '\t' + str(getattr(args, arg)), debug=debug)


def logging_computing(tracker, loss, performance, _input, lr):
    tracker_update_performance(tracker, loss, performance, _input.size(0))
    tracker['computing_time'].update(time.time() - tracker['end_data_time'])
    tracker['start_sync_time'] = time.time()
    tracker['learning_rate'].update(lr)


def logging_sync_time(tracker):
    tracker['sync_time'].update(time.time() - tracker['start_sync_time'])


def logging_load_time(tracker):
    tracker['load_time'].update(time.time() - tracker['start_load_time'])


def logging_globally(tracker, start_global_time):
    tracker['global_time'].update(time.time() - start_global_time)


def logging_display_training(args, tracker):
    log_info = ('Epoch: {epoch:.3f}. Local index: {local_index}. '
                'Load: {load:.3f}s | Data: {data:.3f}s | '
                'Computing: {computing_time:.3f}s | Sync: {sync_time:.3f}s | '
                'Global: {global_time:.3f}s | Loss: {loss:.4f} | '
                'top1: {top1:.4f} | top5: {top5:.4f} | '
                'learning_rate: {lr:.4f} | rounds_comm: {rounds_comm}').format(
        epoch=args.epoch_,
        local_index=args.local_index,
        load=tracker['load_time'].avg,
        data=tracker['data_time'].avg,
        computing_time=tracker['computing_time'].avg,
        sync_time=tracker['sync_time'].avg,
        global_time=tracker['global_time'].avg,
        loss=tracker['losses'].avg,
        top1=tracker['top1'].avg,
        top5=tracker['top5'].avg,
        lr=tracker['learning_rate'].val,
        rounds_comm=args.rounds_comm)
    log('Process {}: '.format(args.graph.rank) + log_info, debug=args.debug)


def logging_display_val(args, performance, mode, personal=False):
    if mode == 'test':
        prefix = 'Test at personal model at batch' if personal else 'Test at batch'
        log('{}: {}. Epoch: {}. Process: {}. Prec@1: {:.3f} Prec@5: {:.3f} '
            'Loss: {:.3f} Comm: {}'.format(
                prefix, args.local_index, args.epoch, args.graph.rank,
                performance[0], performance[1], performance[2],
                args.rounds_comm),
            debug=args.debug)
    else:
        p0 = 'Personal' if personal else 'Global'
        p1 = 'validation' if mode == 'validation' else 'train'
        log('{} performance for {} at batch: {}. Epoch: {}. Process: {}. '
            'Prec@1: {:.3f} Prec@5: {:.3f} Loss: {:.3f} Comm: {}'.format(
                p0, p1, args.local_index, args.epoch, args.graph.rank,
                performance[0], performance[1], performance[2],
                args.rounds_comm),
            debug=args.debug)


def logging_display_test_summary(args, debug=True):
    log('best accuracy for rank {} at local index {} '
        '(best epoch {:.3f}, current epoch {:.3f}): {}.'.format(
            args.graph.rank, args.local_index,
            args.best_epoch[-1] if len(args.best_epoch) != 0 else 0.0,
            args.epoch_, args.best_prec1), debug=debug)


def tracker_update_performance(tracker, loss, performance, size):
    tracker['losses'].update(
        loss.item() if hasattr(loss, 'item') else float(loss), size)
    if len(performance) == 2:
        tracker['top5'].update(performance[1], size)
    tracker['top1'].update(performance[0], size)
    return tracker


# reference-spelled alias (`logging.py` update_performancec_tracker)
update_performancec_tracker = tracker_update_performance


def update_performance_per_class(tracker, acc, count, classes):
    for a, n, c in zip(acc, count, classes):
        tracker[c.item()].update(a.item(), n.item())
    return tracker
