"""HTTP gateway (FastAPI) exposing the node to local clients and the web app."""
