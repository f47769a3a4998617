"""Fixture ASGI user app (factory convention)."""


def make_app():
    from fastapi import FastAPI, Request

    app = FastAPI()

    @app.post("/double")
    async def double(request: Request):
        body = await request.json()
        return {"doubled": body["x"] * 2}

    return app
