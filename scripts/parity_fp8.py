"""fp8-vs-bf16 accuracy parity study (BASELINE config 5
uses fp8 local training).  Trains the headline federation twice on the SAME
learnable synthetic data for an equal number of rounds — once with bf16
compute, once with fp8 conv GEMMs — and prints the Global-Accuracy curves
side by side.

    python scripts/parity_fp8.py [rounds] [spu]
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault('HETEROFL_SYNTHETIC_MODE', 'learnable')

import torch

from heterofl_amd.config import default_config
from heterofl_amd.control import process_control, CONTROL_FIELDS
from heterofl_amd.data import fetch_dataset, split_dataset
from heterofl_amd.fed import FedRunner
from heterofl_amd.logger import Logger
from heterofl_amd.models import make_model
from heterofl_amd.utils import process_dataset, make_optimizer, make_scheduler


def run(dtype, rounds, spu, eval_every):
    cfg = default_config()
    control = '1_100_0.1_iid_fix_a1-e1_bn_1_1'
    cfg['control'] = dict(zip(CONTROL_FIELDS, control.split('_')))
    cfg['control_name'] = control
    cfg['data_name'] = 'CIFAR10'
    cfg['model_name'] = 'resnet18'
    cfg['device'] = 'cuda:0' if torch.cuda.is_available() else 'cpu'
    cfg['engine'] = 'batched'
    cfg['compute_dtype'] = dtype
    cfg['metric_name'] = {'train': {'Local': ['Local-Loss', 'Local-Accuracy']},
                          'test': {'Local': ['Local-Loss', 'Local-Accuracy'],
                                   'Global': ['Global-Loss',
                                              'Global-Accuracy']}}
    process_control(cfg)
    torch.manual_seed(0)
    ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=spu * 100)
    process_dataset(ds, cfg)
    torch.manual_seed(7)
    data_split, label_split = split_dataset(ds, 100, 'iid', 10)
    torch.manual_seed(0)
    model = make_model(cfg).to(cfg['device'])
    opt = make_optimizer(model, cfg['lr'], cfg)
    sch = make_scheduler(opt, cfg)
    logger = Logger(None)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt,
                       logger=logger)
    curve = []
    for ep in range(1, rounds + 1):
        logger.safe(True)
        runner.train_round(ep)
        if ep % eval_every == 0 or ep == rounds:
            tm = runner.stats()
            runner.test(tm, ep)
            curve.append((ep, logger.mean['test/Global-Accuracy'],
                          logger.mean['train/Local-Loss']))
            print(f'[{dtype}] round {ep}: global-acc '
                  f'{curve[-1][1]:.2f}  local-loss {curve[-1][2]:.3f}',
                  flush=True)
        import warnings
        with warnings.catch_warnings():
            warnings.filterwarnings('ignore', message='.*lr_scheduler.step.*')
            sch.step()
        logger.safe(False)
        logger.reset()
    return curve


def main():
    rounds = int(sys.argv[1]) if len(sys.argv) > 1 else 60
    spu = int(sys.argv[2]) if len(sys.argv) > 2 else 100
    eval_every = max(rounds // 6, 1)
    bf16 = run('bfloat16', rounds, spu, eval_every)
    fp8 = run('fp8', rounds, spu, eval_every)
    print('\n=== fp8 vs bf16 parity (equal rounds, same data/seed) ===')
    print(f'{"round":>6} {"bf16_acc":>9} {"fp8_acc":>9} {"delta":>7} '
          f'{"bf16_loss":>10} {"fp8_loss":>9}')
    for (ep, a1, l1), (_, a2, l2) in zip(bf16, fp8):
        print(f'{ep:6d} {a1:9.2f} {a2:9.2f} {a2 - a1:7.2f} {l1:10.3f} '
              f'{l2:9.3f}')


if __name__ == '__main__':
    main()
