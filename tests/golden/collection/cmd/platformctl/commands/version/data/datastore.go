/*
Copyright 2021.

Licensed under the Apache License, Version 2.0 (the "License");
you may not use this file except in compliance with the License.
You may obtain a copy of the License at

    http://www.apache.org/licenses/LICENSE-2.0

Unless required by applicable law or agreed to in writing, software
distributed under the License is distributed on an "AS IS" BASIS,
WITHOUT WARRANTIES OR CONDITIONS OF ANY KIND, either express or implied.
See the License for the specific language governing permissions and
limitations under the License.
*/

package data

import (
	"github.com/spf13/cobra"

	cmdversion "github.com/acme/platform/cmd/platformctl/commands/version"

	"github.com/acme/platform/apis/data"
)

// NewDataStoreSubCommand creates a new command and adds it to its
// parent command.
func NewDataStoreSubCommand(parentCommand *cobra.Command) {
	versionCmd := &cmdversion.VersionSubCommand{
		Name:         "datastore",
		Description:  "Manage the data store component",
		VersionFunc:  VersionDataStore,
		SubCommandOf: parentCommand,
	}

	versionCmd.Setup()
}

func VersionDataStore(v *cmdversion.VersionSubCommand) error {
	apiVersions := make([]string, len(data.DataStoreGroupVersions()))

	for i, groupVersion := range data.DataStoreGroupVersions() {
		apiVersions[i] = groupVersion.Version
	}

	versionInfo := cmdversion.VersionInfo{
		CLIVersion:  cmdversion.CLIVersion,
		APIVersions: apiVersions,
	}

	return versionInfo.Display()
}
