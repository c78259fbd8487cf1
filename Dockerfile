# Server image: the C++ modelxd registry + modelx-s3d object server.
# (CPU-only — the servers never touch GPUs; the GPU data plane lives in the
# client library, which ships with the ROCm client image, Dockerfile.dl.)
FROM ubuntu:22.04 AS build
RUN apt-get update && apt-get install -y --no-install-recommends \
    g++ make libssl-dev zlib1g-dev && rm -rf /var/lib/apt/lists/*
WORKDIR /src
COPY Makefile ./
COPY core ./core
RUN make servers

FROM ubuntu:22.04
RUN apt-get update && apt-get install -y --no-install-recommends \
    libssl3 zlib1g wget ca-certificates && rm -rf /var/lib/apt/lists/*
COPY --from=build /src/bin/modelxd /src/bin/modelx-s3d /app/
EXPOSE 8080 9000
ENTRYPOINT ["/app/modelxd"]
CMD ["--listen", ":8080"]
