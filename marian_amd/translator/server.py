"""marian-server: WebSocket translation service
(ref: src/command/marian_server.cpp + 3rd_party/simple-websocket-server).

Implemented on fastapi/uvicorn (wheelhouse) with a plain-websocket endpoint
at '/translate' mirroring the reference's message-in -> translation-out
protocol, plus an HTTP POST /translate convenience endpoint.
"""
import sys

from ..common.logging_utils import create_loggers, log_info
from .translate import TranslateService


from pydantic import BaseModel


class TranslateRequest(BaseModel):
    text: str


def build_app(service: TranslateService):
    from fastapi import FastAPI, WebSocket, WebSocketDisconnect

    app = FastAPI(title="marian_amd server")
    Req = TranslateRequest

    @app.websocket("/translate")
    async def ws_translate(ws: WebSocket):
        await ws.accept()
        try:
            while True:
                text = await ws.receive_text()
                await ws.send_text(service.run(text))
        except WebSocketDisconnect:
            pass

    @app.post("/translate")
    def http_translate(req: Req):
        return {"translation": service.run(req.text)}

    @app.get("/health")
    def health():
        return {"status": "ok"}

    return app


def main_server(argv=None):
    from ..common.config_parser import ConfigParser, MODE_SERVER
    parser = ConfigParser(MODE_SERVER)
    opts = parser.parse_options(argv if argv is not None else sys.argv[1:],
                                validate=False)
    create_loggers(opts.get("log"), None, opts.get("log-level", "info"),
                   opts.get("quiet", False))
    service = TranslateService(opts)
    app = build_app(service)
    import uvicorn
    port = int(opts.get("port", 8080))
    log_info("marian-server listening on port %d", port)
    uvicorn.run(app, host="0.0.0.0", port=port, log_level="warning")
    return 0
