"""C20 flag-system parity: every flag the reference parser exposes
(/root/reference/src/torchgems/parser.py:29-143) must be accepted with
the same spelling and value semantics, so a torchgems command line is a
drop-in for our entry points."""

from mpi4dl_amd.parser import get_parser, parse_csv_ints

REFERENCE_FLAGS = [
    "--verbose", "--batch-size", "--parts", "--split-size",
    "--num-spatial-parts", "--spatial-size", "--times", "--image-size",
    "--num-epochs", "--num-layers", "--num-filters", "--num-classes",
    "--balance", "--halo-D2", "--fused-layers", "--local-DP",
    "--slice-method", "--app", "--datapath", "--enable-master-comm-opt",
    "--num-workers",
]


def test_reference_flags_accepted():
    opts = {
        a for action in get_parser()._actions for a in action.option_strings
    }
    missing = [f for f in REFERENCE_FLAGS if f not in opts]
    assert not missing, f"reference flags not accepted: {missing}"


def test_reference_command_line_parses():
    # a realistic torchgems SP+GEMS command line, verbatim spelling
    args = get_parser().parse_args([
        "--batch-size", "4", "--parts", "2", "--split-size", "3",
        "--num-spatial-parts", "4,2", "--spatial-size", "2",
        "--times", "2", "--image-size", "1024", "--num-epochs", "1",
        "--num-layers", "18", "--num-filters", "416", "--num-classes", "10",
        "--balance", "3,1,1", "--halo-D2", "--fused-layers", "2",
        "--local-DP", "2", "--slice-method", "vertical", "--app", "3",
        "--enable-master-comm-opt", "--num-workers", "0", "--verbose",
    ])
    assert args.batch_size == 4 and args.parts == 2 and args.split_size == 3
    assert parse_csv_ints(args.num_spatial_parts) == [4, 2]
    assert args.spatial_size == 2 and args.times == 2
    assert parse_csv_ints(args.balance) == [3, 1, 1]
    assert args.halo_d2 is True and args.fused_layers == 2
    assert args.local_DP == 2 and args.slice_method == "vertical"
    assert args.enable_master_comm_opt is True and args.verbose is True


def test_csv_parsing():
    assert parse_csv_ints("4") == [4]
    assert parse_csv_ints("4,2,1") == [4, 2, 1]
    assert parse_csv_ints(None) is None
