from gridllm_amd.server.app import main

main()
