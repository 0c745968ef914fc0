"""Subprocess entry for SDK tests: serve the event server or the engine
query server on a given port, reading storage config from the inherited
PIO_* environment (sqlite file shared with the test process)."""

import sys


def main():
    kind, port = sys.argv[1], int(sys.argv[2])
    if kind == "event":
        from predictionio_amd.server.eventserver import create_app
        app = create_app()
    else:
        from predictionio_amd.server.queryserver import (
            ServerConfig, create_app,
        )
        app = create_app(ServerConfig(engine_factory=sys.argv[3]))
    import uvicorn
    uvicorn.run(app, host="127.0.0.1", port=port, log_level="error")


if __name__ == "__main__":
    main()
