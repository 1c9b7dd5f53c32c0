"""Small utilities shared by the bench and the distributed layer."""

import json
import os
import sys


def env_rank():
    """(rank, world_size, local_rank) from torchrun env, defaulting to
    single-process."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", str(rank)))
    return rank, world, local


def print_json_line(obj):
    sys.stdout.write(json.dumps(obj) + "\n")
    sys.stdout.flush()


def validate_circuit(state, sbox, num_inputs, bit=None):
    """Ground-truth check: evaluates the circuit DAG on every input pattern
    and compares with the S-box table. `bit` limits the check to one output
    bit; None checks all wired outputs."""
    outputs = state.outputs
    for x in range(1 << num_inputs):
        got = state.eval(x)
        want = sbox[x]
        for b in range(8):
            if outputs[b] < 0:
                continue
            if bit is not None and b != bit:
                continue
            if ((got >> b) & 1) != ((want >> b) & 1):
                return False
    return True
