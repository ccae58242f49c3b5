"""Example script-file operator.

The engine invokes this as `python3 train.py --params '<json>'` once
per contiguous client shard (the reference's operator contract,
base_operator.py:12-53 — see docs/USAGE.md §3).  It simulates local
training for `actor_simulation_num` virtual devices and reports
per-device counts by writing result.json in `actor_save_dir`.
"""

import json
import os
import sys


def main() -> int:
    params = json.loads(sys.argv[sys.argv.index("--params") + 1])
    n = params["actor_simulation_num"]
    lo, hi = params["client_range"]
    rnd = params["current_round"]
    extra = json.loads(params["operator"]["operator_params"] or "{}")

    # ... your per-device work goes here; this demo just records that
    # every device in [lo, hi) "trained" for the round ...
    failed = int(extra.get("fail_per_shard", 0))
    failed = min(failed, n)

    with open(os.path.join(params["actor_save_dir"], "result.json"), "w") as f:
        json.dump({"success": n - failed, "failed": failed,
                   "round": rnd, "client_range": [lo, hi]}, f)
    return 0


if __name__ == "__main__":
    sys.exit(main())
