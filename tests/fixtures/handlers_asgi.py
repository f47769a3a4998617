"""Fixture ASGI user app (factory convention)."""


def make_app():
    from fastapi import FastAPI, Request

    app = FastAPI()

    @app.post("/double")
    async def double(request: Request):
        body = await request.json()
        return {"doubled": body["x"] * 2}

    return app


def make_ws_app():
    """Realtime fixture: FastAPI app with a websocket echo route."""
    from fastapi import FastAPI, WebSocket

    app = FastAPI()

    @app.websocket("/ws")
    async def ws_echo(ws: WebSocket):
        await ws.accept()
        while True:
            try:
                msg = await ws.receive_text()
            except Exception:
                break
            await ws.send_text(f"echo:{msg}")

    return app
