#!/usr/bin/env python3
"""Extract the SC2 game-version -> (base build, data version) table into
assets/sc2_versions.json.

The table is Blizzard's published build info
(https://github.com/Blizzard/s2client-proto/blob/master/buildinfo/versions.json),
mirrored in the reference's vendored pysc2 (`pysc2/run_configs/lib.py:36`).
Like actions.json / static_data.json it is a game-data constant, not code —
the version string in a replay's metadata must map to the exact Base<build>
binary directory and -dataVersion hash or SC2 refuses to load the replay.
"""
import ast
import json
import os
import re
import sys

REF = '/root/reference/distar/pysc2/run_configs/lib.py'
OUT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   'distar_amd', 'assets', 'sc2_versions.json')


def main():
    with open(REF) as f:
        src = f.read()
    entries = re.findall(r'Version\("([\d.]+)",\s*(\d+),\s*"([0-9A-F]+)",\s*None\)',
                         src)
    table = {gv: {'base_build': int(bb), 'data_version': dv}
             for gv, bb, dv in entries}
    assert len(table) >= 60, f'only {len(table)} versions parsed'
    with open(OUT, 'w') as f:
        json.dump(table, f, indent=1, sort_keys=True)
    print(f'wrote {len(table)} versions to {OUT}')


if __name__ == '__main__':
    main()
