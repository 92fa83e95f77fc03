"""Process entry: `python -m selkies_amd` / the `selkies` console script.

Mirrors the reference lifecycle (SURVEY.md §3.1 __main__.py): build the
supervisor, start the selected streaming mode, install SIGTERM/SIGHUP
handlers that unwind identically to Ctrl-C, run the loop.
"""

from __future__ import annotations

import asyncio
import logging
import signal
import sys


def main(argv=None) -> int:
    from .settings import load_settings
    from .stream_server import CentralizedStreamServer

    settings = load_settings(argv=argv)
    logging.basicConfig(
        level=logging.DEBUG if settings.debug else logging.INFO,
        format="%(asctime)s %(name)s %(levelname)s %(message)s")

    async def run():
        server = CentralizedStreamServer(settings)
        await server.start()
        stop = asyncio.Event()
        loop = asyncio.get_running_loop()
        for sig in (signal.SIGTERM, signal.SIGINT, signal.SIGHUP):
            try:
                loop.add_signal_handler(sig, stop.set)
            except (NotImplementedError, OSError):
                pass
        try:
            await stop.wait()
        finally:
            await server.stop()

    try:
        asyncio.run(run())
    except KeyboardInterrupt:
        pass
    return 0


if __name__ == "__main__":
    sys.exit(main())
