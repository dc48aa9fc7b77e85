"""Elastic launcher — ``python -m bagua_amd.distributed.run``.

Thin front of torchelastic (``torch.distributed.run``) adding the bagua
flags (reference: bagua/distributed/run.py:180-414, 578-639): rendezvous
and restart semantics are delegated entirely to torchelastic, which works
unchanged on ROCm; bagua flags are exported as BAGUA_* env vars for the
workers. The reference's --enable_bagua_net flag is intentionally gone:
bagua-net was a TCP inter-node NCCL plugin and this build targets
single-node xGMI (SURVEY.md §5 "Drop bagua-net").
"""

import os
import sys

from torch.distributed import run as torch_run


def get_args_parser():
    parser = torch_run.get_args_parser()
    parser.description = (
        "bagua_amd elastic launcher (torchelastic + BAGUA_* env)")
    parser.add_argument(
        "--bagua_service_port", "--bagua-service-port",
        type=int, default=-1,
        help="fixed port for the autotune service (default: auto)")
    parser.add_argument(
        "--default_bucket_size", "--default-bucket-size",
        type=int, default=32 * 1024 * 1024,
        help="fusion bucket size in bytes (default 32 MiB)")
    parser.add_argument(
        "--autotune_level", "--autotune-level", type=int, default=0,
        help="0 = off, 1 = bucket-size/hierarchy autotuning")
    parser.add_argument(
        "--autotune_max_samples", type=int, default=60)
    parser.add_argument(
        "--autotune_sampling_confidence_time", type=float, default=5.0)
    parser.add_argument(
        "--autotune_warmup_time", type=float, default=30.0)
    parser.add_argument(
        "--is_output_autotune_log", type=int, default=0)
    parser.add_argument(
        "--report_metrics", action="store_true")
    return parser


def parse_args(args):
    return get_args_parser().parse_args(args)


def set_bagua_env(args, current_env=None):
    env = current_env if current_env is not None else os.environ
    env["BAGUA_DEFAULT_BUCKET_SIZE"] = str(args.default_bucket_size)
    env["BAGUA_AUTOTUNE"] = str(args.autotune_level)
    env["BAGUA_AUTOTUNE_MAX_SAMPLES"] = str(args.autotune_max_samples)
    env["BAGUA_AUTOTUNE_SAMPLING_CONFIDENCE_TIME_S"] = str(
        args.autotune_sampling_confidence_time)
    env["BAGUA_AUTOTUNE_WARMUP_TIME_S"] = str(args.autotune_warmup_time)
    env["BAGUA_IS_OUTPUT_AUTOTUNE_LOG"] = str(args.is_output_autotune_log)
    if args.bagua_service_port > 0:
        env["BAGUA_SERVICE_PORT"] = str(args.bagua_service_port)
    if args.report_metrics:
        env["BAGUA_REPORT_METRICS"] = "1"


def run(args):
    set_bagua_env(args)
    # keep in-tree bagua_amd importable from worker scripts in other dirs
    os.environ["PYTHONPATH"] = os.pathsep.join(
        [os.getcwd()] + [p for p in
                         os.environ.get("PYTHONPATH", "").split(os.pathsep)
                         if p])
    torch_run.run(args)


def main(args=None):
    args = parse_args(args if args is not None else sys.argv[1:])
    run(args)


if __name__ == "__main__":
    main()
