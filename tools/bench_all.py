"""Scripted runs of the five BASELINE.json configs -> one JSON report.

SURVEY.md §7 step 7. Each config shells out to the driver-contract
bench.py (or main.py for the CPU plumbing config) and collects the JSON
line; multi-GPU configs are launched under torchrun with one rank per
visible GPU (they degrade to however many GPUs the box has).

python tools/bench_all.py [--out report.json] [--steps 30]
"""
import argparse
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run(cmd, timeout=1200):
    env = dict(os.environ)
    env['PYTHONPATH'] = REPO
    env.setdefault('MASTER_ADDR', '127.0.0.1')
    r = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                       text=True, timeout=timeout)
    line = None
    for ln in reversed(r.stdout.strip().splitlines()):
        ln = ln.strip()
        if ln.startswith('{') and '"metric"' in ln:
            line = json.loads(ln)
            break
    return {'cmd': ' '.join(cmd), 'rc': r.returncode, 'result': line,
            'stderr_tail': r.stderr[-400:] if r.returncode else ''}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--out', type=str, default=None)
    ap.add_argument('--steps', type=int, default=30)
    ap.add_argument('--warmup', type=int, default=5)
    args = ap.parse_args()
    py = sys.executable
    import torch
    ngpu = torch.cuda.device_count() if torch.cuda.is_available() else 0

    report = {'configs': {}}

    # 1: CPU plumbing (tiny synthetic train through the real CLI)
    r = subprocess.run(
        [py, 'main.py', '--train-flag', '--synthetic', '--synthetic-size',
         '4', '--gpu-no', '-1', '--batch-size', '2', '--end-epoch', '1',
         '--num-workers', '0', '--imsize', '128', '--num-stack', '1',
         '--hourglass-inch', '16', '--save-path', '/tmp/bench_all_cpu/'],
        cwd=REPO, env={**os.environ, 'PYTHONPATH': REPO},
        capture_output=True, text=True, timeout=900)
    report['configs']['1_cpu_plumbing'] = {
        'rc': r.returncode,
        'ok': r.returncode == 0 and
              os.path.exists('/tmp/bench_all_cpu/check_point_1.pth')}

    if ngpu >= 1:
        s, w = str(args.steps), str(args.warmup)
        # 2: bf16 inference, 1 GPU (100 FPS target)
        report['configs']['2_infer_bf16_1gpu'] = run(
            [py, 'bench.py', '--mode', 'infer', '--batch-size', '1',
             '--steps', '100', '--warmup', '20', '--graph'])
        # 5: fp8 MFMA + hipGraph decode, batch 8
        report['configs']['5_infer_fp8_graph_b8'] = run(
            [py, 'bench.py', '--mode', 'infer', '--batch-size', '8',
             '--steps', '50', '--warmup', '10', '--graph', '--fp8'])
        # 3: DDP bf16 training (as many GPUs as the box has)
        if ngpu > 1:
            base = [py, '-m', 'torch.distributed.run', '--nnodes=1',
                    f'--nproc-per-node={ngpu}', '--master-addr',
                    '127.0.0.1', '--master-port', '29741']
            report['configs']['3_train_ddp'] = run(
                base + ['bench.py', '--gpus', str(ngpu), '--steps', s,
                        '--warmup', w])
            report['configs']['4_train_big_ddp'] = run(
                base[:-2] + ['--master-port', '29742', 'bench.py',
                             '--gpus', str(ngpu), '--steps', '20',
                             '--warmup', w, '--num-stack', '2',
                             '--increase-ch', '128'])
        else:
            report['configs']['3_train_1gpu'] = run(
                [py, 'bench.py', '--steps', s, '--warmup', w])
            report['configs']['4_train_big_1gpu'] = run(
                [py, 'bench.py', '--steps', '20', '--warmup', w,
                 '--num-stack', '2', '--increase-ch', '128'])

    out = json.dumps(report, indent=2)
    print(out)
    if args.out:
        with open(args.out, 'w') as f:
            f.write(out)


if __name__ == '__main__':
    main()
