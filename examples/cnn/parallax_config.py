"""Flags -> ParallaxConfig translation for the CNN benchmark example (the
reference's per-example parallax_config.py convention,
examples/lm1b/parallax_config.py:19-80)."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import parallax_amd


def build_config(args):
    """args: the example driver's argparse namespace."""
    ckpt = parallax_amd.CheckPointConfig(
        ckpt_dir=getattr(args, 'ckpt_dir', None),
        save_ckpt_steps=getattr(args, 'save_ckpt_steps', None),
        save_ckpt_secs=getattr(args, 'save_ckpt_secs', None))
    profile = parallax_amd.ProfileConfig(
        profile_dir=getattr(args, 'profile_dir', None),
        profile_steps=getattr(args, 'profile_steps', None),
        profile_worker=getattr(args, 'profile_worker', 0))
    comm = parallax_amd.CommunicationConfig(
        ps_config=parallax_amd.PSConfig(
            local_aggregation=getattr(args, 'local_aggregation', True)))
    comm.dense_comm_dtype = 'bf16'
    return parallax_amd.Config(
        run_option=getattr(args, 'run_option', 'HYBRID'),
        average_sparse=getattr(args, 'average_sparse', False),
        redirect_path=getattr(args, 'redirect_path', None),
        search_partitions=getattr(args, 'search_partitions', False),
        export_graph_path=getattr(args, 'export_graph_path', None),
        communication_config=comm,
        ckpt_config=ckpt,
        profile_config=profile)
