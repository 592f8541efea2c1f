#!/bin/bash
# Dense S-SGD baseline launcher (reference parity: the horovod/dense
# baseline scripts): same flags, compression=none.
compression=none density=1 exec "$(dirname "$0")/dist.sh" "$@"
