"""Split a libffm data file into N per-rank shards by row hash — parity
with the reference's offline sharder (/root/reference/data/proc_file_split.py
feeding the per-rank loader distributed_algo_abst.h:97-100).

Usage: python tools/proc_file_split.py data.csv N  ->  data_1.csv..data_N.csv
"""

import sys
import zlib


def main():
    path, n = sys.argv[1], int(sys.argv[2])
    outs = [open(f"{path.rsplit('.', 1)[0]}_{i + 1}.csv", "w")
            for i in range(n)]
    with open(path) as f:
        for i, line in enumerate(f):
            outs[zlib.crc32(str(i).encode()) % n].write(line)
    for o in outs:
        o.close()


if __name__ == "__main__":
    main()
