"""CLI entry: dispatch train vs evaluate (reference /root/reference/main.py).

python main.py --train-flag [flags]   -> distributed multi-GPU training
python main.py [flags]                -> evaluation (restores arch flags from
                                         the checkpoint's argument.pickle)
"""

import time
from datetime import timedelta

from real_time_helmet_detection_amd.config import get_arguments
from real_time_helmet_detection_amd.parallel import distributed_device_train
from real_time_helmet_detection_amd.engine import single_device_evaluate

if __name__ == '__main__':
    args = get_arguments()

    tictoc = time.time()
    if args.train_flag:
        distributed_device_train(args)
    else:
        single_device_evaluate(args)
    print('%s: finished in %s'
          % (time.ctime(), timedelta(seconds=time.time() - tictoc)))
