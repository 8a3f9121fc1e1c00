"""storage-initializer entrypoint (init-container contract).

Reference parity: python/storage-initializer/scripts/initializer-entrypoint —
argv = src_uri dest [src_uri dest ...]; writes /mnt/models.
Run: python -m kserve_amd.storage.initializer <src> <dest> [...]
"""

import sys

from kserve_amd.logging import configure_logging, logger
from kserve_amd.storage import Storage


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    configure_logging()
    if len(argv) < 2 or len(argv) % 2 != 0:
        logger.error("usage: initializer <src_uri> <dest> [<src_uri> <dest>...]")
        return 1
    Storage.download_files(argv)
    return 0


if __name__ == "__main__":
    sys.exit(main())
