"""Helper algorithm library (reference: e2/src/main/scala/.../e2/)."""
