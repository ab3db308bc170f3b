"""Circuit-breaker monitor endpoint (reference parity:
services/utils/circuit_breaker_monitor.py:15-140 — REST on :9091 exposing
and resetting breakers). FastAPI app factory; mounted standalone or into
the dashboard app."""

from __future__ import annotations

from .circuit_breaker import all_breakers


def build_breaker_app():
    from fastapi import FastAPI, HTTPException

    app = FastAPI(title="circuit-breaker-monitor")

    @app.get("/breakers")
    async def list_breakers():
        return {name: br.status() for name, br in all_breakers().items()}

    @app.get("/breakers/{name}")
    async def one(name: str):
        brs = all_breakers()
        if name not in brs:
            raise HTTPException(404, f"no breaker '{name}'")
        return brs[name].status()

    @app.post("/breakers/{name}/reset")
    async def reset(name: str):
        brs = all_breakers()
        if name not in brs:
            raise HTTPException(404, f"no breaker '{name}'")
        brs[name].reset()
        return brs[name].status()

    @app.post("/breakers/reset_all")
    async def reset_all():
        for br in all_breakers().values():
            br.reset()
        return {"reset": len(all_breakers())}

    return app


def main(port: int = 9091):
    import uvicorn

    uvicorn.run(build_breaker_app(), host="127.0.0.1", port=port,
                log_level="warning")


if __name__ == "__main__":
    main()
