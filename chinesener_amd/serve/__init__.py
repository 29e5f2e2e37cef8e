"""L6 serving: export layout, hipGraph-captured inference, gRPC server/client."""
