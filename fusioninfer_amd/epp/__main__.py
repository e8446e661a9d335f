from fusioninfer_amd.epp.router_server import main

if __name__ == "__main__":
    main()
