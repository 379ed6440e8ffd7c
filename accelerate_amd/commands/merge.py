"""`accelerate-amd merge-weights` — merge sharded FSDP checkpoints into a
single full state dict (reference: commands/merge.py, fsdp_utils.py:462)."""



def merge_command(args):
    from ..parallel.fsdp_io import merge_fsdp_weights

    merge_fsdp_weights(args.checkpoint_directory, args.output_path, safe_serialization=not args.unsafe_serialization)
    print(f"Merged weights written to {args.output_path}")


def add_parser(subparsers):
    parser = subparsers.add_parser("merge-weights", help="Merge sharded FSDP checkpoints into one file")
    parser.add_argument("checkpoint_directory", help="Directory containing per-rank sharded checkpoints")
    parser.add_argument("output_path", help="Where to write the merged weights")
    parser.add_argument("--unsafe_serialization", action="store_true", help="Write pytorch .bin instead of safetensors")
    parser.set_defaults(func=merge_command)
    return parser
